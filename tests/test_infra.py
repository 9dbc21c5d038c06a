"""Infra tests: graph IR, MSequential, profiler, TConfig, env, hwinfo
(reference graph_test, msequential_example, archiver_test, config plumbing)."""

import json
import time

import torch

from tnn_amd import nn as tnn
from tnn_amd.nn.graph import Graph, GraphBuilder, GraphExecutor, wrap_sequential
from tnn_amd.utils import TConfig, Profiler, EventType, env_get, hwinfo
from tnn_amd.utils.env import EnvLoader


def test_graph_topo_and_exec():
    b = GraphBuilder("g")
    b.add_layer(tnn.Dense(4, 8, True, "d1"), ["input"], "h")
    b.add_layer(tnn.Activation("relu", "r"), ["h"], "hr")
    b.add_layer(tnn.Dense(8, 2, True, "d2"), ["hr"], "output")
    g = b.build()
    ex = GraphExecutor(g)
    x = torch.randn(3, 4)
    (out,) = ex.forward({"input": x}, ["output"])
    assert out.shape == (3, 2)
    # backward through the executor
    x = torch.randn(3, 4, requires_grad=True)
    (out,) = ex.forward({"input": x}, ["output"])
    ex.backward([out], [torch.ones_like(out)])
    assert x.grad is not None
    ex.zero_grads()
    assert all(p.grad is None for p in g.parameters())


def test_graph_config_roundtrip():
    b = GraphBuilder("g2")
    b.add_layer(tnn.Dense(4, 4, True, "d"), ["input"], "output")
    g = b.build()
    cfg = g.get_config()
    g2 = Graph.from_config(cfg)
    assert g2.get_config() == cfg


def test_wrap_sequential_single_edge():
    """The reference's standard shape: one OpNode wrapping the model
    (example_models.hpp:49-71)."""
    m = (tnn.LayerBuilder((4,)).dense(8).activation("relu").dense(2).build())
    g = wrap_sequential(m)
    out = g({"input": torch.randn(2, 4)})["output"]
    assert out.shape == (2, 2)


def test_msequential_joins():
    for join in ["add", "mul", "concat"]:
        branches = [tnn.Sequential([tnn.Dense(4, 6, True, f"{join}_b{i}")])
                    for i in range(2)]
        ms = tnn.MSequential(branches, join=join)
        y = ms(torch.randn(3, 4))
        expect = 12 if join == "concat" else 6
        assert y.shape == (3, expect)
        assert ms.output_shape((4,)) == (expect,)
        cfg = ms.get_config()
        from tnn_amd.nn.layer import layer_from_config
        assert layer_from_config(cfg).get_config() == cfg


def test_profiler_spans_merge_trace(tmp_path):
    p1 = Profiler("rank0")
    p1.start()
    with p1.span(EventType.COMPUTE, "fwd"):
        time.sleep(0.01)
    p2 = Profiler("rank1")
    p2.start()
    with p2.span(EventType.COMMUNICATION, "send"):
        time.sleep(0.005)
    p1.merge(Profiler.from_dict(p2.to_dict()))
    assert len(p1.events) == 2
    assert p1.summary()["fwd"] >= 0.01
    trace = tmp_path / "t.json"
    p1.export_chrome_trace(str(trace))
    data = json.loads(trace.read_text())
    assert len(data["traceEvents"]) == 2


def test_tconfig_roundtrip(tmp_path):
    c = TConfig()
    c.set("model.name", "wrn").set("model.depth", 16).set("lr", 0.1)
    c.set("flags.amp", True)
    assert c.get("model.depth") == 16
    assert c.get("missing", 42) == 42
    assert c.has("flags.amp")
    path = tmp_path / "c.json"
    c.save(str(path))
    c2 = TConfig.from_file(str(path))
    assert c2 == c


def test_env_loader(tmp_path, monkeypatch):
    envfile = tmp_path / ".env"
    envfile.write_text("FOO_X=17\n# comment\nBAR_Y=hello\n")
    monkeypatch.delenv("FOO_X", raising=False)
    EnvLoader.load(str(envfile))
    assert env_get("FOO_X", int) == 17
    assert env_get("BAR_Y") == "hello"
    assert env_get("NOPE_Z", int, 5) == 5


def test_hwinfo_probe():
    info = hwinfo.HardwareInfo.probe()
    assert info.cpu_count >= 1
    assert isinstance(info.gpus, list)


def test_hwinfo_topology_parse():
    """xGMI matrix parsing (both rocm-smi output forms) + chain order."""
    from tnn_amd.utils.hwinfo import HardwareInfo, _parse_topo_matrix
    table = (
        "        GPU0  GPU1  GPU2\n"
        "GPU0    0     XGMI  PCIE\n"
        "GPU1    XGMI  0     XGMI\n"
        "GPU2    PCIE  XGMI  0\n")
    links = _parse_topo_matrix(table)
    assert links[(0, 1)] == "XGMI" and links[(0, 2)] == "PCIE"
    pairform = ("(Topology) Link type between DRM devices 0 and 1: XGMI\n"
                "(Topology) Link type between DRM devices 0 and 2: PCIE\n")
    links2 = _parse_topo_matrix(pairform)
    assert links2[(1, 0)] == "XGMI" and links2[(2, 0)] == "PCIE"

    hw = HardwareInfo(cpu_count=8)
    from tnn_amd.utils.hwinfo import GPUInfo
    hw.gpus = [GPUInfo(i, "MI355X", 0, 256) for i in range(3)]
    hw.link_type = links
    assert hw.xgmi_peers(1) == [0, 2]
    order = hw.pipeline_order()
    assert order == [0, 1, 2]  # chain follows xGMI neighbors


def test_visualize_profiler_entrypoint(tmp_path):
    """visualizers/visualize_profiler.py renders a Profiler chrome trace."""
    import subprocess, sys, os, time
    from tnn_amd.utils.profiler import Profiler, EventType
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    prof = Profiler("test")
    prof.start()
    with prof.span(EventType.COMPUTE, "fwd"):
        time.sleep(0.002)
    with prof.span(EventType.COMMUNICATION, "allreduce"):
        time.sleep(0.001)
    prof.stop()
    trace = tmp_path / "trace.json"
    prof.export_chrome_trace(str(trace))
    r = subprocess.run(
        [sys.executable,
         os.path.join(root, "visualizers", "visualize_profiler.py"),
         str(trace)], capture_output=True, text=True, timeout=60)
    assert r.returncode == 0, r.stderr
    assert "fwd" in r.stdout and "span" in r.stdout
