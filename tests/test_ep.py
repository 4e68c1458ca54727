"""Expert parallelism: 2-rank gloo test — EP MoE forward must equal the
single-process full-expert reference (parallel/ep.py, BASELINE config 4)."""
import os
import sys

import torch
import torch.multiprocessing as mp

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _ep_worker(rank: int, world: int, port: int, q):
    try:
        os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                          LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                          MASTER_PORT=str(port), AURORA_EP="1")
        sys.path.insert(0, ROOT)
        import torch
        from aurora_amd.models.config import get_spec
        from aurora_amd.models.llama import MoEMLP, _FullWidthMLP
        from aurora_amd.parallel import dist as pdist
        from aurora_amd.parallel import ep as pep

        pdist.init_distributed(tp_size=world, backend="gloo")
        assert pep.ep_enabled() and pep.ep_size() == world
        spec = get_spec("smoke-moe")
        torch.manual_seed(0)
        # reference: ALL experts full-width, deterministic weights
        E = spec.n_experts
        gus = [torch.randn(2 * spec.intermediate_size, spec.hidden_size)
               * spec.hidden_size ** -0.5 for _ in range(E)]
        dns = [torch.randn(spec.hidden_size, spec.intermediate_size)
               * spec.intermediate_size ** -0.5 for _ in range(E)]
        router_w = torch.randn(E, spec.hidden_size) * spec.hidden_size ** -0.5

        m = MoEMLP(spec, torch.float32)
        assert m.ep_size == world and isinstance(m.experts[0], _FullWidthMLP)
        e_local = E // world
        with torch.no_grad():
            m.router.weight.copy_(router_w)
            for le in range(e_local):
                ge = rank * e_local + le
                m.experts[le].gate_up.copy_(gus[ge])
                m.experts[le].down.copy_(dns[ge])

        torch.manual_seed(42)
        x = torch.randn(13, spec.hidden_size)
        y = m(x)

        # single-process reference computed identically on every rank
        logits = (x @ router_w.t()).float()
        w, sel = logits.topk(spec.experts_per_token, dim=-1)
        w = w.softmax(-1)
        ref = torch.zeros_like(x)
        import torch.nn.functional as F
        for t in range(x.shape[0]):
            for j in range(spec.experts_per_token):
                e = int(sel[t, j])
                gu = x[t] @ gus[e].t()
                g, u = gu.chunk(2)
                ref[t] += w[t, j] * ((F.silu(g) * u) @ dns[e].t())
        err = (y - ref).abs().max().item() / max(ref.abs().max().item(), 1e-6)
        pdist.destroy()
        q.put((rank, err))
    except Exception as e:  # noqa: BLE001
        import traceback
        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


def test_ep_moe_matches_reference():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_ep_worker, args=(r, 2, 29561, q))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = [q.get(timeout=240) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, err in outs:
        assert not isinstance(err, str), err
        assert err < 1e-4, (rank, err)


def test_ep_moe_four_ranks():
    """EP=4 on gloo: uneven routing across 4 owners still matches the
    reference computation (the Mixtral TP=4 shape, BASELINE config 4)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_ep_worker, args=(r, 4, 29571, q))
             for r in range(4)]
    for p in procs:
        p.start()
    outs = [q.get(timeout=300) for _ in range(4)]
    for p in procs:
        p.join(timeout=60)
    for rank, err in outs:
        assert not isinstance(err, str), err
        assert err < 1e-4, (rank, err)
