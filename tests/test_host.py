"""Host-side tests (no GPU): library loads + exports every declared symbol,
topology YAML parsing (range expressions, contiguity), bf16 helpers."""
import ctypes
import os
import re

import numpy as np
import pytest

import cake_amd
from cake_amd import CakeHipError, topology_node_range

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

TOPO = """\
# cake-style topology (sharding/topology.rs:134-169)
linux_server_1:
  host: 192.168.1.2:10128
  description: NVIDIA Titan X Pascal (12GB)
  layers:
    - model.layers.0-15
linux_server_2:
  host: 192.168.1.3:10128
  layers:
    - model.layers.16-31
odd_node:
  host: 192.168.1.4:10128
  layers:
    - model.layers.40
    - model.layers.42
single:
  host: 192.168.1.5:10128
  layers:
    - model.layers.7
"""


def test_library_loads_and_reports_gfx950():
    assert "gfx950" in cake_amd.build_info()


def test_header_symbols_all_exported():
    """Every cake_hip_* function declared in include/cake_hip.h must be an
    exported symbol of the built library (tier framing §3: the C-ABI library
    loads and exports every declared entry point)."""
    hdr = open(os.path.join(REPO, "include", "cake_hip.h")).read()
    names = set(re.findall(r"\b(cake_hip_\w+)\s*\(", hdr))
    assert len(names) >= 15
    for n in sorted(names):
        assert hasattr(cake_amd._lib, n), f"symbol {n} not exported"


def test_topology_range_expansion():
    # "model.layers.0-15" expands per topology.rs:142-166
    assert topology_node_range(TOPO, "linux_server_1") == (0, 16)
    assert topology_node_range(TOPO, "linux_server_2") == (16, 32)
    assert topology_node_range(TOPO, "single") == (7, 8)


def test_topology_rejects_noncontiguous():
    with pytest.raises(CakeHipError, match="contiguous"):
        topology_node_range(TOPO, "odd_node")


def test_topology_missing_node():
    with pytest.raises(CakeHipError, match="not in topology"):
        topology_node_range(TOPO, "nope")


def test_topology_invalid_range_order():
    bad = "n:\n  layers:\n    - model.layers.5-2\n"
    with pytest.raises(CakeHipError):
        topology_node_range(bad, "n")


def test_quantize_bf16_matches_rne():
    from tests.helpers import quantize_bf16
    import torch
    x = np.random.default_rng(0).standard_normal(4096).astype(np.float32)
    ours = quantize_bf16(x)
    ref = torch.tensor(x).to(torch.bfloat16).float().numpy()
    assert np.array_equal(ours, ref)


def test_engine_requires_gpu_no_silent_fallback():
    """On a box with no GPU the engine must fail loudly, not fall back to
    CPU (tier framing §3)."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present")
    with pytest.raises(CakeHipError):
        cake_amd.Engine(dict(
            model_type="llama", hidden_size=64, intermediate_size=128,
            vocab_size=128, num_hidden_layers=1, num_attention_heads=4,
            num_key_value_heads=2, head_dim=16), max_seq=32)


def test_engine_rejects_bad_config():
    # config parsing precedes any GPU call, so these are CPU-testable
    import json as _json
    with pytest.raises(_json.JSONDecodeError):
        cake_amd.Engine("not json at all {", max_seq=32)  # wrapper-side
    with pytest.raises(CakeHipError, match="missing required"):
        cake_amd.Engine(dict(model_type="llama", hidden_size=64,
                             num_hidden_layers=1), max_seq=32)


def test_engine_rejects_bad_layer_range():
    with pytest.raises(CakeHipError, match="layer range"):
        cake_amd.Engine(dict(
            model_type="llama", hidden_size=64, intermediate_size=128,
            vocab_size=128, num_hidden_layers=2, num_attention_heads=4,
            num_key_value_heads=2, head_dim=16), layer_lo=1, layer_hi=1)
