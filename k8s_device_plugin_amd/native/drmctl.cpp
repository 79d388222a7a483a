// Raw DRM_IOCTL_AMDGPU_INFO shim (C++/pybind11).
//
// MI355X-native replacement for the reference's libdrm/libdrm_amdgpu cgo
// bindings (reference: internal/pkg/amdgpu/amdgpu.go:21-27,86-101,358-448,
// 551-563).  Instead of linking libdrm_amdgpu we issue the kernel UAPI
// ioctls directly on /dev/dri/renderD*|card*: firmware/feature versions
// (AMDGPU_INFO_FW_VERSION), device info (AMDGPU_INFO_DEV_INFO: family,
// device id, CU count), VRAM totals (AMDGPU_INFO_VRAM_GTT), and a liveness
// probe (open + one info ioctl), so the plugin has zero userspace GPU
// library dependencies.
//
// Build: g++ -O2 -shared -fPIC (no HIP, no libdrm) — see native/build.py.

#include <cerrno>
#include <cstdint>
#include <cstring>
#include <fcntl.h>
#include <string>
#include <sys/ioctl.h>
#include <unistd.h>

#include <drm/amdgpu_drm.h>

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

namespace py = pybind11;

namespace {

struct Fd {
    int fd;
    explicit Fd(const std::string &path) : fd(::open(path.c_str(), O_RDWR | O_CLOEXEC)) {}
    ~Fd() { if (fd >= 0) ::close(fd); }
    bool ok() const { return fd >= 0; }
};

int amdgpu_info(int fd, uint32_t query, void *out, uint32_t out_size,
                uint32_t fw_type = 0) {
    struct drm_amdgpu_info req;
    std::memset(&req, 0, sizeof(req));
    req.return_pointer = reinterpret_cast<uint64_t>(out);
    req.return_size = out_size;
    req.query = query;
    if (query == AMDGPU_INFO_FW_VERSION)
        req.query_fw.fw_type = fw_type;
    return ::ioctl(fd, DRM_IOCTL_AMDGPU_INFO, &req);
}

// The 10 firmware blocks the reference queries (amdgpu.go:416-445), same
// label set so the labeller emits identical firmware label keys.
struct FwBlock { const char *label; uint32_t fw_type; };
const FwBlock kFwBlocks[] = {
    {"VCE", AMDGPU_INFO_FW_VCE},
    {"UVD", AMDGPU_INFO_FW_UVD},
    {"MC", AMDGPU_INFO_FW_GMC},
    {"ME", AMDGPU_INFO_FW_GFX_ME},
    {"PFP", AMDGPU_INFO_FW_GFX_PFP},
    {"CE", AMDGPU_INFO_FW_GFX_CE},
    {"RLC", AMDGPU_INFO_FW_GFX_RLC},
    {"MEC", AMDGPU_INFO_FW_GFX_MEC},
    {"SMC", AMDGPU_INFO_FW_SMC},
    {"SDMA0", AMDGPU_INFO_FW_SDMA},
};

py::dict query_firmware(const std::string &dev_path) {
    Fd dev(dev_path);
    if (!dev.ok())
        throw std::runtime_error("cannot open " + dev_path + ": " + std::strerror(errno));

    py::dict feat_versions, fw_versions;
    for (const auto &blk : kFwBlocks) {
        struct drm_amdgpu_info_firmware fw;
        std::memset(&fw, 0, sizeof(fw));
        if (amdgpu_info(dev.fd, AMDGPU_INFO_FW_VERSION, &fw, sizeof(fw),
                        blk.fw_type) == 0) {
            feat_versions[blk.label] = fw.feature;
            fw_versions[blk.label] = fw.ver;
        } else {
            feat_versions[blk.label] = 0;
            fw_versions[blk.label] = 0;
        }
    }
    py::dict out;
    out["feature"] = feat_versions;
    out["firmware"] = fw_versions;
    return out;
}

py::dict query_device_info(const std::string &dev_path) {
    Fd dev(dev_path);
    if (!dev.ok())
        throw std::runtime_error("cannot open " + dev_path + ": " + std::strerror(errno));

    struct drm_amdgpu_info_device info;
    std::memset(&info, 0, sizeof(info));
    if (amdgpu_info(dev.fd, AMDGPU_INFO_DEV_INFO, &info, sizeof(info)) != 0)
        throw std::runtime_error("AMDGPU_INFO_DEV_INFO failed on " + dev_path +
                                 ": " + std::strerror(errno));

    py::dict out;
    out["device_id"] = info.device_id;
    out["chip_rev"] = info.chip_rev;
    out["external_rev"] = info.external_rev;
    out["pci_rev"] = info.pci_rev;
    out["family"] = info.family;
    out["num_shader_engines"] = info.num_shader_engines;
    out["cu_active_number"] = info.cu_active_number;
    out["max_engine_clock_khz"] = info.max_engine_clock;
    out["max_memory_clock_khz"] = info.max_memory_clock;
    out["vram_type"] = info.vram_type;
    out["vram_bit_width"] = info.vram_bit_width;
    return out;
}

py::dict query_vram(const std::string &dev_path) {
    Fd dev(dev_path);
    if (!dev.ok())
        throw std::runtime_error("cannot open " + dev_path + ": " + std::strerror(errno));

    struct drm_amdgpu_info_vram_gtt vram;
    std::memset(&vram, 0, sizeof(vram));
    if (amdgpu_info(dev.fd, AMDGPU_INFO_VRAM_GTT, &vram, sizeof(vram)) != 0)
        throw std::runtime_error("AMDGPU_INFO_VRAM_GTT failed on " + dev_path +
                                 ": " + std::strerror(errno));
    py::dict out;
    out["vram_size"] = vram.vram_size;
    out["vram_cpu_accessible_size"] = vram.vram_cpu_accessible_size;
    out["gtt_size"] = vram.gtt_size;
    return out;
}

bool dev_functional(const std::string &dev_path) {
    // open + one info ioctl: a stronger liveness probe than the reference's
    // bare open (amdgpu.go:390-399) while staying kernel-only.
    Fd dev(dev_path);
    if (!dev.ok())
        return false;
    struct drm_amdgpu_info_device info;
    std::memset(&info, 0, sizeof(info));
    return amdgpu_info(dev.fd, AMDGPU_INFO_DEV_INFO, &info, sizeof(info)) == 0;
}

} // namespace

PYBIND11_MODULE(_drmctl, m) {
    m.doc() = "raw DRM_IOCTL_AMDGPU_INFO bindings (no libdrm)";
    m.def("query_firmware", &query_firmware,
          "10 firmware/feature versions via AMDGPU_INFO_FW_VERSION",
          py::arg("dev_path"));
    m.def("query_device_info", &query_device_info,
          "family/device-id/CU info via AMDGPU_INFO_DEV_INFO",
          py::arg("dev_path"));
    m.def("query_vram", &query_vram,
          "VRAM/GTT sizes via AMDGPU_INFO_VRAM_GTT", py::arg("dev_path"));
    m.def("dev_functional", &dev_functional,
          "liveness: open + DEV_INFO ioctl succeeds", py::arg("dev_path"));
}
