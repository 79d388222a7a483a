"""debugfs firmware parser tests (reference: TestParseDebugFSFirmwareInfo,
amdgpu_test.go:171-224)."""

from k8s_device_plugin_amd.topology.firmware import (
    parse_debugfs_firmware,
    parse_debugfs_firmware_text,
)

SAMPLE = """\
VCE feature version: 0, firmware version: 0x00000000
UVD feature version: 0, firmware version: 0x00000000
MC feature version: 0, firmware version: 0x00000000
ME feature version: 55, firmware version: 0x00000040
PFP feature version: 55, firmware version: 0x0000004f
CE feature version: 55, firmware version: 0x0000002b
RLC feature version: 1, firmware version: 0x00000049
MEC feature version: 55, firmware version: 0x000001a1
MEC2 feature version: 55, firmware version: 0x000001a1
SOS feature version: 0, firmware version: 0x00161a63
ASD feature version: 0, firmware version: 0x2116276c
SMC feature version: 0, firmware version: 0x00362500
SDMA0 feature version: 52, firmware version: 0x000000a6
SDMA1 feature version: 52, firmware version: 0x000000a6
VCN feature version: 0, firmware version: 0x0110901c
"""


def test_parse_debugfs_firmware_text():
    feat, fw = parse_debugfs_firmware_text(SAMPLE)
    assert len(feat) == 15 and len(fw) == 15
    assert feat["MEC"] == 55
    assert fw["MEC"] == 0x1A1
    assert fw["SMC"] == 0x362500
    assert feat["SDMA1"] == 52
    assert fw["VCN"] == 0x0110901C


def test_parse_debugfs_firmware_file(tmp_path):
    p = tmp_path / "amdgpu_firmware_info"
    p.write_text(SAMPLE)
    feat, fw = parse_debugfs_firmware(str(p))
    assert feat["PFP"] == 55 and fw["PFP"] == 0x4F


def test_parse_debugfs_missing():
    assert parse_debugfs_firmware("/nonexistent/path") == ({}, {})


def test_ignores_garbage():
    feat, fw = parse_debugfs_firmware_text("hello\nnot a fw line\n")
    assert feat == {} and fw == {}
