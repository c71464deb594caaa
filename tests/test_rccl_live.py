"""Live-communicator de-risk tests.

At world=1 on a real GPU, ``torch.distributed`` with the nccl backend IS a
live RCCL communicator: every ``all_reduce`` executes for real (identity
value, real collective), including inside hipGraph capture — exactly the
capture-sensitive path that gates config 5 (SURVEY.md §7 hard-part 4).
These run under plain pytest on the GPU box (no torchrun needed: the test
initializes the group itself over 127.0.0.1).
"""

import json
import os
import socket
import subprocess
import sys

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.fixture()
def nccl_world1():
    """A real RCCL communicator with world_size=1 on cuda:0."""
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    assert not torch.distributed.is_initialized()
    torch.cuda.set_device(0)
    torch.distributed.init_process_group(
        "nccl", rank=0, world_size=1,
        init_method=f"tcp://127.0.0.1:{_free_port()}",
    )
    yield
    torch.distributed.destroy_process_group()


class TestRcclWorld1:
    def test_all_reduce_executes(self, nccl_world1):
        from flowml import dist

        assert dist.is_initialized() and dist.get_world_size() == 1
        t = torch.full((1024,), 3.0, device="cuda:0")
        dist.all_reduce_(t)
        assert torch.all(t == 3.0)
        dist.all_reduce_(t, op="max")
        dist.all_reduce_(t, op="min")
        assert torch.all(t == 3.0)

    def test_all_reduce_inside_hipgraph(self, nccl_world1):
        """RCCL collective captured in a hipGraph and replayed (config 5
        gating risk: capture-safe RCCL was never proven with a live
        communicator in round 1)."""
        from flowml import dist

        x = torch.ones(1 << 20, device="cuda:0")
        acc = torch.zeros(1, device="cuda:0")

        def work():
            dist.all_reduce_(x)
            acc.add_(x.sum())

        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                work()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        acc.zero_()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, capture_error_mode="thread_local"):
            work()
        acc.zero_()
        for _ in range(5):
            g.replay()
        torch.cuda.synchronize()
        assert float(acc.item()) == 5 * x.numel()

    def test_fused_allreduce_forced_inside_graph(self, nccl_world1):
        """FusedAllReduce with the world==1 short-circuit disabled, under
        capture: pack → RCCL all_reduce → unpack all inside the graph."""
        from flowml.dist import FusedAllReduce

        a = torch.zeros(513, device="cuda:0")
        b = torch.zeros(31, 7, device="cuda:0")
        f = FusedAllReduce([a, b])

        def work():
            f.reduce(force=True)

        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            work()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, capture_error_mode="thread_local"):
            work()
        a.fill_(2.0)
        b.fill_(-1.5)
        g.replay()
        torch.cuda.synchronize()
        assert torch.all(a == 2.0) and torch.all(b == -1.5)

    def test_feature_estimators_under_live_comm(self, nccl_world1):
        """Sharded-stat all-reduces in the feature estimators run on the
        real communicator (world=1 identity) without device mismatches."""
        from flowml.feature import MaxAbsScaler, MinMaxScaler, StandardScaler
        from flowml.frame import FlowFrame

        x = torch.randn(10_000, 8, device="cuda:0")
        df = FlowFrame({"vec": x})
        ss = StandardScaler(inputCol="vec", outputCol="o").fit(df)
        assert np.allclose(ss.mean.numpy(), x.mean(0).double().cpu().numpy(), atol=1e-6)
        mm = MinMaxScaler(inputCol="vec", outputCol="o").fit(df)
        assert np.allclose(mm.originalMax.numpy(), x.max(0).values.double().cpu().numpy())
        MaxAbsScaler(inputCol="vec", outputCol="o").fit(df)

    def test_round2_estimators_under_live_comm(self, nccl_world1):
        """Round-2 estimators whose stat reduces must be NCCL-device-safe:
        RobustScaler (dist quantile machinery), UnivariateFeatureSelector
        (on-device contingency/moment reduces), GMM/GLR flat buffers."""
        from flowml.feature import RobustScaler, UnivariateFeatureSelector
        from flowml.frame import FlowFrame
        from flowml.regression import GeneralizedLinearRegression

        r = np.random.default_rng(3)
        X = torch.from_numpy(r.normal(size=(20_000, 5)).astype(np.float32)).to("cuda:0")
        yk = torch.from_numpy(r.integers(0, 3, 20_000).astype(np.float64)).to("cuda:0")
        df = FlowFrame({"features": X, "label": yk}, device="cuda:0")
        rs = RobustScaler(inputCol="features", outputCol="s").fit(df)
        assert rs.range.shape == (5,)
        sel = UnivariateFeatureSelector(
            featuresCol="features", outputCol="s", labelCol="label",
            featureType="continuous", labelType="categorical",
            selectionThreshold=2.0).fit(df)
        assert len(sel.selectedFeatures) == 2
        # categorical/categorical chi2 path (device contingency reduce)
        Xc = torch.from_numpy(r.integers(0, 4, (20_000, 3)).astype(np.float32)).to("cuda:0")
        dfc = FlowFrame({"features": Xc, "label": yk}, device="cuda:0")
        sel2 = UnivariateFeatureSelector(
            featuresCol="features", outputCol="s", labelCol="label",
            featureType="categorical", labelType="categorical",
            selectionThreshold=1.0).fit(dfc)
        assert len(sel2.selectedFeatures) == 1
        yp = torch.from_numpy(
            r.poisson(2.0, 20_000).astype(np.float64)).to("cuda:0")
        GeneralizedLinearRegression(family="poisson").fit(
            FlowFrame({"features": X, "label": yp}, device="cuda:0"))


# NOTE: a world=2 communicator on ONE GPU was tried and is NOT possible:
# RCCL (like NCCL) rejects two ranks on the same device ("duplicate GPU",
# ncclInvalidUsage) — verified on the MI355X box. Multi-rank RCCL semantics
# are covered by the gloo world=2/4/8 suites; the live-RCCL coverage above
# (world=1, incl. in-graph) is the maximum a 1-GPU lease allows.


class TestTorchrunWorld1:
    def test_bench_mlp_under_torchrun_rccl(self):
        """torchrun --nproc-per-node 1 bench.py: a real RCCL communicator
        initializes, the fused grad all_reduce is captured inside the
        bench hipGraph, and the JSON line reports both facts."""
        if not torch.cuda.is_available():
            pytest.skip("needs GPU")
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        env = dict(os.environ)
        env.pop("RANK", None), env.pop("WORLD_SIZE", None)
        r = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node=1", "--master-addr=127.0.0.1",
             f"--master-port={_free_port()}", "bench.py", "--gpus", "1",
             "--model", "mlp", "--steps", "5", "--warmup", "2",
             "--rows", "200000"],
            cwd=repo, env=env, capture_output=True, text=True, timeout=600,
        )
        assert r.returncode == 0, r.stdout + r.stderr
        line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
        out = json.loads(line)
        assert out["n_gpus"] == 1
        assert out["config"]["rccl_live"] is True
        assert out["config"]["hipgraph"] is True, (
            "hipGraph capture fell back to eager under a live RCCL communicator"
        )
