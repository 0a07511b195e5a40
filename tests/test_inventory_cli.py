"""amd-smi CLI fallback parser against REAL captured output
(tests/fixtures/*, captured from an MI355X box via gpurun this round)."""
import json
import os

from gpu_docker_api_amd.parallel.inventory import AmdSmiInventory

FIXTURES = os.path.join(os.path.dirname(os.path.abspath(__file__)), "fixtures")


def test_parse_real_amdsmi_output():
    data = json.load(open(os.path.join(FIXTURES, "amdsmi_list.json")))
    static = json.load(open(os.path.join(FIXTURES, "amdsmi_static.json")))
    static_by_gpu = {int(g["gpu"]): g for g in static["gpu_data"]}
    gpus = AmdSmiInventory.parse_cli_output(data, static_by_gpu)
    assert len(gpus) == 1
    g = gpus[0]
    assert g.uuid == "8dff75a3-0000-1000-80ff-65c1a6d56ef6"
    assert g.bdf == "0000:23:00.0"
    assert g.name == "AMD Instinct MI355 OAM"
    # 294896 MB HBM3E
    assert g.vram_total == 294896 * 1024**2


def test_parse_without_static_enrichment():
    data = json.load(open(os.path.join(FIXTURES, "amdsmi_list.json")))
    gpus = AmdSmiInventory.parse_cli_output(data)
    assert len(gpus) == 1
    assert gpus[0].vram_total == 288 * 1024**3  # spec default
    assert gpus[0].index == 0


def test_parse_multi_gpu_shape():
    data = [
        {"gpu": i, "bdf": f"0000:{0x20 + i:02x}:00.0", "uuid": f"u-{i}"} for i in range(8)
    ]
    gpus = AmdSmiInventory.parse_cli_output(data)
    assert [g.index for g in gpus] == list(range(8))
    assert all(g.uuid == f"u-{g.index}" for g in gpus)
