"""DistributedOptimizer correctness at world_size 2 (gloo): bucketed
all-reduce averaging, unused-parameter robustness, gradient
accumulation (backward_passes_per_step)."""

import unittest

from sparkdl import HorovodRunner


def _averaging_main():
    import torch
    import sparkdl.torch as hvd
    hvd.init()
    r = hvd.rank()
    torch.manual_seed(0)
    model = torch.nn.Sequential(
        torch.nn.Linear(4, 8), torch.nn.Linear(8, 2))
    hvd.broadcast_parameters(model, root_rank=0)
    opt = hvd.DistributedOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.0))  # lr=0: inspect grads

    x = torch.full((2, 4), float(r + 1))
    opt.zero_grad()
    model(x).sum().backward()
    opt.step()  # synchronizes + averages into p.grad

    # reference: average of per-rank grads = grads of mean input batch
    model_ref = torch.nn.Sequential(
        torch.nn.Linear(4, 8), torch.nn.Linear(8, 2))
    model_ref.load_state_dict(model.state_dict())
    g_sum = None
    for rr in range(hvd.size()):
        model_ref.zero_grad()
        model_ref(torch.full((2, 4), float(rr + 1))).sum().backward()
        gs = [p.grad.clone() for p in model_ref.parameters()]
        g_sum = gs if g_sum is None else [a + b for a, b in zip(g_sum, gs)]
    ok = all(
        torch.allclose(p.grad, g / hvd.size(), atol=1e-6)
        for p, g in zip(model.parameters(), g_sum))
    return bool(ok)


def _unused_param_main():
    import torch
    import sparkdl.torch as hvd
    hvd.init()

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.used = torch.nn.Linear(4, 4)
            self.unused = torch.nn.Linear(4, 4)

        def forward(self, x):
            return self.used(x)

    torch.manual_seed(0)
    m = M()
    hvd.broadcast_parameters(m, root_rank=0)
    opt = hvd.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.1))
    opt.zero_grad()
    m(torch.randn(2, 4)).sum().backward()
    opt.step()  # must not hang on the never-fired bucket
    return True


def _grad_accum_main():
    import torch
    import sparkdl.torch as hvd
    hvd.init()
    torch.manual_seed(0)
    p = torch.nn.Parameter(torch.zeros(4))
    opt = hvd.DistributedOptimizer(
        torch.optim.SGD([p], lr=1.0), backward_passes_per_step=2)
    opt.zero_grad()
    # two backward passes accumulate, then one averaged step
    (p * (hvd.rank() + 1.0)).sum().backward()
    (p * (hvd.rank() + 1.0)).sum().backward()
    opt.step()
    # grad per rank = 2*(rank+1); average over ranks {1,2} = 3
    expected = -3.0
    return bool(torch.allclose(p.detach(),
                               torch.full((4,), expected), atol=1e-6))


class DistributedOptimizerTestCase(unittest.TestCase):
    def test_grad_averaging(self):
        self.assertTrue(HorovodRunner(np=-2).run(_averaging_main))

    def test_unused_params_no_hang(self):
        self.assertTrue(HorovodRunner(np=-2).run(_unused_param_main))

    def test_backward_passes_per_step(self):
        self.assertTrue(HorovodRunner(np=-2).run(_grad_accum_main))


if __name__ == "__main__":
    unittest.main()


def _dist_fused_adamw_main():
    """The bench wiring: DistributedOptimizer wrapping FusedAdamW
    (CPU reference path), vs single-process AdamW on averaged grads."""
    import torch
    import sparkdl.torch as hvd
    import sparkdl.ops as ops
    hvd.init()
    torch.manual_seed(0)
    model = torch.nn.Linear(8, 8)
    hvd.broadcast_parameters(model, root_rank=0)
    ref = torch.nn.Linear(8, 8)
    ref.load_state_dict(model.state_dict())

    opt = hvd.DistributedOptimizer(
        ops.FusedAdamW(model.parameters(), lr=1e-2, weight_decay=0.01))
    ropt = torch.optim.AdamW(ref.parameters(), lr=1e-2, weight_decay=0.01)

    for step in range(3):
        opt.zero_grad()
        x = torch.full((4, 8), float(hvd.rank() + step + 1))
        model(x).sum().backward()
        opt.step()

        # reference: average gradient over ranks
        ropt.zero_grad()
        gs = None
        for rr in range(hvd.size()):
            tmp = torch.nn.Linear(8, 8)
            tmp.load_state_dict(ref.state_dict())
            tmp(torch.full((4, 8), float(rr + step + 1))).sum().backward()
            g = [p.grad for p in tmp.parameters()]
            gs = g if gs is None else [a + b for a, b in zip(gs, g)]
        for p, g in zip(ref.parameters(), gs):
            p.grad = g / hvd.size()
        ropt.step()

    return all(torch.allclose(a, b, atol=1e-5)
               for a, b in zip(model.parameters(), ref.parameters()))


class DistFusedAdamWTestCase(unittest.TestCase):
    def test_bench_wiring_cpu(self):
        self.assertTrue(HorovodRunner(np=-2).run(_dist_fused_adamw_main))


def _world4_entry(rank, world, port):
    import os
    import torch
    import torch.distributed as dist
    os.environ.update({"RANK": str(rank), "WORLD_SIZE": str(world),
                       "LOCAL_RANK": str(rank),
                       "MASTER_ADDR": "127.0.0.1",
                       "MASTER_PORT": str(port),
                       "SPARKDL_USE_GPU": "0"})
    import sparkdl.torch as hvd
    hvd.init()
    torch.manual_seed(77)
    model = torch.nn.Linear(16, 4)
    opt = hvd.DistributedOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.1),
        named_parameters=model.named_parameters())
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    g = torch.Generator().manual_seed(500 + rank)
    for _ in range(3):
        x = torch.randn(8, 16, generator=g)
        opt.zero_grad()
        model(x).pow(2).mean().backward()
        opt.step()
    s = float(sum(p.detach().sum() for p in model.parameters()))
    sums = hvd.allgather_object(round(s, 5))
    hvd.shutdown()
    assert len(set(sums)) == 1, sums


def test_world4_gloo_distopt():
    """4-rank gloo gang: the bucketed DistributedOptimizer keeps all
    ranks' parameters identical (rehearses the np=4 GPU shape without
    hardware)."""
    import multiprocessing as mp
    from sparkdl.engine.rendezvous import free_port
    port = free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_world4_entry, args=(r, 4, port))
             for r in range(4)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0, p.exitcode


def test_bucket_mb_env_override(monkeypatch):
    """SPARKDL_BUCKET_MB drives the gradient bucket capacity."""
    monkeypatch.setenv("SPARKDL_BUCKET_MB", "1")
    import importlib
    from sparkdl.parallel import distributed_optimizer as dopt
    importlib.reload(dopt)
    try:
        assert dopt._DEFAULT_BUCKET_MB == 1.0
    finally:
        monkeypatch.delenv("SPARKDL_BUCKET_MB")
        importlib.reload(dopt)
