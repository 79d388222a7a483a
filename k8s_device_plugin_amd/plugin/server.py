"""DevicePlugin v1beta1 gRPC servicer.

Parity with the reference's AMDGPUPlugin (reference:
internal/pkg/plugin/plugin.go:41-397):
  - ListAndWatch walks sysfs once, caches the device map for Allocate, sends
    the initial list, then re-sends with refreshed health on every heartbeat;
  - Allocate serves purely from the in-memory cache: /dev/kfd always plus
    /dev/dri/card<N> + /dev/dri/renderD<M> per requested device — zero
    syscalls on the pod-admission path;
  - GetPreferredAllocation delegates to the hive-aware best-effort policy;
  - heterogeneous nodes bucket devices by '<compute>_<memory>' and each
    resource's plugin serves only its bucket.

Deviation (improvement) from the reference: on ListAndWatch stream loss the
reference os.Exit(1)s so the DaemonSet restarts the pod (plugin.go:322-324).
We instead notify the lifecycle manager, which re-registers with the kubelet
without killing the process; `exit_on_stream_loss=True` restores the
reference behavior.
"""

from __future__ import annotations

import logging
import os
import threading
from typing import Callable, Dict, List, Optional

from ..allocator import AllocationError, BestEffortPolicy
from ..protos import deviceplugin as dp
from ..protos import metricssvc as ms
from ..topology import (
    GPUDevice,
    KFDTopology,
    SysPaths,
    discover_gpus,
    is_homogeneous,
    simple_health_check,
)
from ..health.exporter import populate_per_gpu_health

log = logging.getLogger(__name__)


class AMDGPUPlugin:
    """One plugin instance per advertised resource (amd.com/<resource>)."""

    def __init__(
        self,
        resource: str = "gpu",
        paths: SysPaths = SysPaths(),
        allocator: Optional[BestEffortPolicy] = None,
        exporter_socket: str = ms.EXPORTER_SOCKET,
        exporter_timeout: float = ms.QUERY_TIMEOUT_S,
        on_stream_lost: Optional[Callable[[], None]] = None,
        exit_on_stream_loss: bool = False,
        cdi_enabled: bool = False,
        cdi_spec_dir: Optional[str] = None,
        prestart_probe: bool = False,
        prestart_deep: bool = False,
        deep_probe_every: int = 0,
        dev_root: str = "/dev",
    ):
        self.resource = resource
        self.cdi_enabled = cdi_enabled
        self.cdi_spec_dir = cdi_spec_dir
        # when enabled, PreStartContainer verifies each requested device
        # answers before the container starts (the reference's
        # PreStartContainer is a no-op and never advertised,
        # plugin.go:219-224)
        self.prestart_probe = prestart_probe
        # additionally run the full MFMA/LDS/HBM probe (with performance
        # floors) before the container starts — catches a GPU whose device
        # node answers but whose matrix pipe or HBM is silently degraded
        self.prestart_deep = prestart_deep
        # every Nth heartbeat, deep-probe each physical GPU and pin the
        # whole GPU (all partitions) Unhealthy on correctness or floor
        # failure; 0 disables (reference has no deep health at all)
        self.deep_probe_every = deep_probe_every
        self._deep_beat = 0
        self._deep_failed: set = set()  # dev_ids that failed the deep probe
        self.dev_root = dev_root
        self.paths = paths
        self.devices: Dict[str, GPUDevice] = {}
        self.allocator = allocator or BestEffortPolicy()
        self.allocator_init_error = False
        self.exporter_socket = exporter_socket
        self.exporter_timeout = exporter_timeout
        self.on_stream_lost = on_stream_lost
        self.exit_on_stream_loss = exit_on_stream_loss
        self._cond = threading.Condition()
        self._heartbeat_gen = 0
        self._latest_list = None  # snapshot shared by all streams per beat
        self._stop = threading.Event()

    # ---- lifecycle ----

    def start(self) -> None:
        """Discover devices and init the allocator (reference: plugin.go:82-91).

        Allocator failure degrades to kubelet-default allocation instead of
        failing startup (plugin.go:86-89).
        """
        topo = KFDTopology.load(self.paths)
        self.devices = discover_gpus(self.paths, topology=topo, strict=False)
        schedulable = [d for d in self.devices.values() if d.kfd_backed]
        try:
            self.allocator.init(schedulable, topology=topo)
        except AllocationError as e:
            log.error(
                "allocator init failed, falling back to kubelet default "
                "allocation: %s", e
            )
            self.allocator_init_error = True

    def stop(self) -> None:
        self._stop.set()
        with self._cond:
            self._cond.notify_all()  # wake streams so they exit

    def refresh_state(self):
        """One shared refresh per heartbeat: a single sysfs walk + a single
        exporter query, whose result every open ListAndWatch stream then
        sends.  Before consolidation each stream independently re-walked
        sysfs and re-dialed the exporter per beat (2 streams = 2 walks);
        the native server already worked this way (native_server.py:79).

        Also mirrors the native heartbeat's device-set tracking: if the
        rediscovered set differs (hot-unplug, partition-mode change) the
        allocator is re-inited so GetPreferredAllocation never serves
        stale groups/weights (the reference has this gap — plugin.go:231
        refreshes the cache but never re-runs allocator Init)."""
        topo = KFDTopology.load(self.paths)
        fresh = discover_gpus(self.paths, topology=topo, strict=False)
        changed = set(fresh) != set(self.devices) or any(
            fresh[i].render_d != self.devices[i].render_d for i in fresh
        )
        self.devices = fresh
        if changed:
            log.info(
                "device set changed (%d devices); re-initializing allocator",
                len(fresh),
            )
            schedulable = [d for d in fresh.values() if d.kfd_backed]
            self.allocator_init_error = False
            try:
                self.allocator.init(schedulable, topology=topo)
            except AllocationError as e:
                log.error(
                    "allocator re-init failed, falling back to kubelet "
                    "default allocation: %s", e
                )
                self.allocator_init_error = True
        self.maybe_deep_check()
        return self.refreshed_device_list(topology=topo)

    # ---- deep GPU health (beyond the reference) ----

    def maybe_deep_check(self) -> None:
        """Deep-probe cadence shared by both serving paths: every Nth
        heartbeat when --deep-probe-every is set (called from the python
        refresh and from the native server's heartbeat push)."""
        if self.deep_probe_every <= 0:
            return
        self._deep_beat += 1
        if self._deep_beat % self.deep_probe_every == 0:
            self._run_deep_check()

    def _hip_ordinal(self, dev: GPUDevice) -> int:
        """HIP device ordinal for a plugin device.

        Primary: exact match of the device's PCI address (dev_id, format
        "dddd:bb:dd:f") against the runtime's per-ordinal PCI bus ids
        ("dddd:bb:dd.f") — robust against ROCR_VISIBLE_DEVICES
        reordering.  Fallback when the probe extension can't enumerate:
        sorted physical dev_id rank (HIP's default PCI enumeration
        order)."""
        try:
            from ..native import load_healthprobe

            mod = load_healthprobe(required=False)
            if mod is not None and hasattr(mod, "pci_bus_ids"):
                want = dev.dev_id.lower().replace(".", ":")
                for ordinal, bus in enumerate(mod.pci_bus_ids()):
                    if bus and bus.lower().replace(".", ":") == want:
                        return ordinal
        except Exception:  # enumeration impossible (no GPU) -> fallback
            pass
        phys = sorted({d.dev_id for d in self.devices.values() if d.kfd_backed})
        return phys.index(dev.dev_id)

    def _run_deep_check(self) -> None:
        """Deep-probe every physical GPU; failures pin the whole GPU
        (all its partitions) Unhealthy until a later probe passes."""
        if not os.path.exists(os.path.join(self.dev_root, "kfd")):
            # fake-sysfs / CPU environment: there is no GPU to probe;
            # skip rather than condemning every (synthetic) device
            log.debug("deep probe skipped: %s/kfd not present", self.dev_root)
            return
        from ..native import NativeExtensionMissing, deep_health_probe

        seen = set()
        for dev in sorted(self.devices.values(), key=lambda d: d.id):
            if not dev.kfd_backed or dev.dev_id in seen:
                continue
            seen.add(dev.dev_id)
            try:
                res = deep_health_probe(device=self._hip_ordinal(dev))
            except NativeExtensionMissing as e:
                # deployment problem, not a device problem: scream, don't flap
                log.error("deep probe unavailable: %s", e)
                return
            except Exception as e:
                log.error("deep probe failed on %s: %s", dev.dev_id, e)
                self._deep_failed.add(dev.dev_id)
                continue
            if res.get("healthy"):
                self._deep_failed.discard(dev.dev_id)
            else:
                log.error(
                    "deep probe UNHEALTHY on %s: violations=%s",
                    dev.dev_id, res.get("floor_violations"),
                )
                self._deep_failed.add(dev.dev_id)

    def heartbeat(self) -> None:
        """Refresh device/health state once and fan it out to all streams."""
        if not self._stop.is_set():
            try:
                latest = self.refresh_state()
            except Exception:
                log.exception("heartbeat refresh failed; keeping last list")
                latest = self._latest_list
        else:
            latest = self._latest_list
        with self._cond:
            self._latest_list = latest
            self._heartbeat_gen += 1
            self._cond.notify_all()

    # ---- DevicePlugin v1beta1 RPCs ----

    def GetDevicePluginOptions(self, request, context):
        opts = dp.DevicePluginOptions()
        if not self.allocator_init_error:
            opts.get_preferred_allocation_available = True
        if self.prestart_probe:
            opts.pre_start_required = True
        return opts

    def render_device_path(self, dev: GPUDevice) -> str:
        return f"{self.dev_root}/dri/renderD{dev.render_d}"

    def PreStartContainer(self, request, context):
        if self.prestart_probe:
            from ..native import load_drmctl

            drm = load_drmctl()
            for dev_id in request.devices_ids:
                dev = self.devices.get(dev_id)
                if dev is None:
                    continue
                path = self.render_device_path(dev)
                ok = os.path.exists(path)
                if ok and drm is not None:
                    import stat

                    # the DRM_IOCTL probe only makes sense on a real device
                    # node (tests use plain files)
                    if stat.S_ISCHR(os.stat(path).st_mode):
                        ok = drm.dev_functional(path)
                if not ok:
                    import grpc

                    log.error("PreStartContainer: device %s (%s) not "
                              "functional", dev_id, path)
                    context.abort(
                        grpc.StatusCode.FAILED_PRECONDITION,
                        f"device {dev_id} failed the pre-start probe",
                    )
                if self.prestart_deep:
                    from ..native import deep_health_probe

                    res = deep_health_probe(device=self._hip_ordinal(dev))
                    if not res.get("healthy"):
                        import grpc

                        log.error(
                            "PreStartContainer: device %s failed the deep "
                            "probe: %s", dev_id, res.get("floor_violations"),
                        )
                        context.abort(
                            grpc.StatusCode.FAILED_PRECONDITION,
                            f"device {dev_id} failed the deep pre-start "
                            f"probe: {res.get('floor_violations')}",
                        )
        return dp.PreStartContainerResponse()

    def _my_devices(self) -> List[GPUDevice]:
        """Devices this resource's plugin advertises."""
        if is_homogeneous(self.devices):
            return list(self.devices.values())
        return [
            d for d in self.devices.values() if d.partition_key == self.resource
        ]

    def _device_list(self, health_default: Optional[str] = None) -> List:
        out = []
        for d in sorted(self._my_devices(), key=lambda x: x.id):
            # a PCI-visible GPU with no kfd topology node (cgroup-masked
            # peer) is not schedulable: advertise it Unhealthy
            health = (health_default or dp.HEALTHY) if d.kfd_backed else dp.UNHEALTHY
            dev = dp.Device(ID=d.id, health=health)
            dev.topology.nodes.add().ID = d.numa_node
            out.append(dev)
        return out

    def refreshed_device_list(self, topology=None) -> List:
        """Device list with freshly evaluated health: node-level kfd scan
        default, exporter per-GPU overrides, unbacked devices pinned
        Unhealthy.  Shared by the Python stream loop and the native
        server's heartbeat push (which passes its already-loaded topology
        to avoid a second sysfs walk per beat)."""
        default = (
            dp.HEALTHY
            if simple_health_check(self.paths, topology=topology)
            else dp.UNHEALTHY
        )
        devs = self._device_list(health_default=default)
        populate_per_gpu_health(
            devs, default, self.exporter_socket, self.exporter_timeout
        )
        # an exporter verdict cannot resurrect a device the kfd topology
        # does not back, nor one whose physical GPU failed the deep probe
        unbacked = {d.id for d in self._my_devices() if not d.kfd_backed}
        deep_failed = {
            d.id
            for d in self._my_devices()
            if d.dev_id and d.dev_id in self._deep_failed
        }
        for dev in devs:
            if dev.ID in unbacked or dev.ID in deep_failed:
                dev.health = dp.UNHEALTHY
        return devs

    def ListAndWatch(self, request, context):
        # re-walk sysfs and refresh the Allocate cache (plugin.go:231)
        self.devices = discover_gpus(self.paths, strict=False)
        log.info("found %d AMD GPU devices", len(self.devices))

        devs = self._device_list()
        # snapshot the heartbeat generation BEFORE the first send: a beat
        # firing while the initial response is in flight must not be lost
        with self._cond:
            gen = self._heartbeat_gen
        yield dp.ListAndWatchResponse(devices=devs)

        while True:
            with self._cond:
                self._cond.wait_for(
                    lambda: self._heartbeat_gen != gen or self._stop.is_set(),
                    timeout=1.0,
                )
                fired = self._heartbeat_gen != gen
                gen = self._heartbeat_gen
            if self._stop.is_set():
                return
            if not context.is_active():
                log.error(
                    "ListAndWatch stream disconnected; triggering re-registration"
                )
                if self.exit_on_stream_loss:
                    os._exit(1)
                if self.on_stream_lost is not None:
                    self.on_stream_lost()
                return
            if not fired:
                continue

            # send the heartbeat's shared snapshot (one walk + one exporter
            # query per beat, not per stream); fall back to computing our
            # own only if a beat fired without a snapshot (direct
            # heartbeat-gen bump in tests)
            with self._cond:
                latest = self._latest_list
            if latest is None:
                latest = self.refreshed_device_list()
            yield dp.ListAndWatchResponse(devices=latest)

    def GetPreferredAllocation(self, request, context):
        import grpc

        response = dp.PreferredAllocationResponse()
        for req in request.container_requests:
            try:
                ids = self.allocator.allocate(
                    list(req.available_deviceIDs),
                    list(req.must_include_deviceIDs),
                    int(req.allocation_size),
                )
            except AllocationError as e:
                log.error("preferred allocation failed: %s", e)
                context.abort(
                    grpc.StatusCode.INVALID_ARGUMENT,
                    f"unable to get preferred allocation list: {e}",
                )
                return response
            response.container_responses.add().deviceIDs.extend(ids)
        return response

    def Allocate(self, request, context):
        response = dp.AllocateResponse()
        for req in request.container_requests:
            car = response.container_responses.add()
            # one /dev/kfd per node, always (plugin.go:368-374)
            kfd = car.devices.add()
            kfd.host_path = kfd.container_path = "/dev/kfd"
            kfd.permissions = "rw"
            for dev_id in req.devices_ids:
                dev = self.devices.get(dev_id)
                if dev is None:
                    log.warning("Allocate: unknown device ID %s", dev_id)
                    continue
                for path in (
                    f"/dev/dri/card{dev.card}",
                    f"/dev/dri/renderD{dev.render_d}",
                ):
                    spec = car.devices.add()
                    spec.host_path = spec.container_path = path
                    spec.permissions = "rw"
                if self.cdi_enabled:
                    from .cdi import cdi_device_name

                    car.cdi_devices.add().name = cdi_device_name(dev_id)
        return response
