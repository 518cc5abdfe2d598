"""Measure real K-FAC factor drift between inverse phases, and simulate
warm-started adaptive block-Jacobi convergence on the captured pairs.

Decides the round-2 eigensolver design (profiles/qdwh_bench.md): if
T = Q_prev^T F_next Q_prev is near-diagonal and the adaptive sweep
count is small, the warm path replaces rocSOLVER syevd for every phase
after the first.

Run on the GPU box:
  python scripts/factor_drift_capture.py > gpurun_out/drift.txt 2>&1
"""

from __future__ import annotations

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def capture_factors(steps_a=100, steps_b=200):
    """Train ResNet-50 on synthetic data; snapshot big factors at two
    consecutive inverse phases (and the init phase for the cold case)."""
    from kfac_amd import KFACPreconditioner
    from kfac_amd.models import resnet50

    device = torch.device('cuda', 0)
    torch.manual_seed(0)
    model = resnet50().to(device)
    model.train()
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    precon = KFACPreconditioner(
        model,
        factor_update_steps=10,
        inv_update_steps=100,
        damping=0.001,
        lr=0.1,
        inv_update_async=False,
    )
    crit = torch.nn.CrossEntropyLoss()
    gen = torch.Generator(device=device).manual_seed(17)

    def batch():
        # FRESH batch per step: with a fixed batch the factors form a
        # commuting family (EMA mixes identity with one constant
        # covariance) and the eigenbasis never rotates — measured
        # off(T) ~ 1e-20, a degenerate best case.  Fresh i.i.d. batches
        # drift the basis through the real mechanism (weight updates).
        x = torch.randn(64, 3, 224, 224, device=device, generator=gen)
        y = torch.randint(0, 1000, (64,), device=device, generator=gen)
        return x, y

    # pick the largest-factor layers
    wanted = {}
    for _, (name, layer) in precon._layers.items():
        n = layer.module.a_factor_shape[0]
        if n >= 1024:
            wanted.setdefault(n, []).append((name, layer))
    for n in wanted:
        wanted[n] = wanted[n][:2]
    snaps: dict[int, dict[str, torch.Tensor]] = {}

    def snap(tag):
        s = {}
        for n, items in wanted.items():
            for name, layer in items:
                f = layer.a_factor
                if isinstance(f, torch.Tensor):
                    s[f'{name}:A{n}'] = f.detach().to(torch.float32).clone()
                g = layer.g_factor
                if isinstance(g, torch.Tensor):
                    s[f'{name}:G{g.shape[0]}'] = (
                        g.detach().to(torch.float32).clone()
                    )
        snaps[tag] = s

    t0 = time.time()
    for step in range(steps_b + 1):
        opt.zero_grad(set_to_none=True)
        x, y = batch()
        with torch.autocast('cuda', dtype=torch.bfloat16):
            loss = crit(model(x), y)
        loss.backward()
        precon.step()
        opt.step()
        if step == 0:
            snap(0)
        elif step == steps_a:
            snap(steps_a)
        elif step == steps_b:
            snap(steps_b)
    torch.cuda.synchronize()
    print(f'training {steps_b} steps took {time.time() - t0:.1f}s')
    return snaps


def block_norms(t, b):
    n = t.size(-1)
    nb = n // b
    m = t[: nb * b, : nb * b].reshape(nb, b, nb, b)
    return torch.sqrt((m * m).sum(dim=(1, 3)))


def simulate_block_jacobi(f_prev, f_next, b=64, target_rel=1e-4, max_sweeps=12):
    """Exact simulation of warm-started adaptive block-Jacobi."""
    n_true = f_prev.size(-1)
    w0, q0 = torch.linalg.eigh(f_prev)
    t = q0.transpose(-1, -2) @ f_next @ q0
    t = 0.5 * (t + t.transpose(-1, -2))
    # pad to a block multiple with decoupled diagonal entries: their
    # off-diagonal coupling is exactly zero, so they never get selected
    # or mixed; this handles ragged sizes (e.g. the fc 2049 = 2048+bias
    # factor, whose excluded boundary row previously stalled the sweep).
    n = ((n_true + b - 1) // b) * b
    if n != n_true:
        tp = t.new_zeros(n, n)
        tp[:n_true, :n_true] = t
        scale = torch.diagonal(t).abs().max()
        tp.diagonal()[n_true:] = scale * torch.linspace(
            2.0, 3.0, n - n_true, device=t.device,
        )
        t = tp
        q0p = q0.new_zeros(n_true, n)
        q0p[:, :n_true] = q0
        q0 = q0p
    tn = torch.linalg.norm(t)
    off0 = torch.linalg.norm(t - torch.diag(torch.diagonal(t)))
    q = q0.clone()
    stats = {'off0_rel': float(off0 / tn), 'rounds': 0, 'pairs': 0}
    nb = n // b
    for sweep in range(max_sweeps):
        off = torch.linalg.norm(t - torch.diag(torch.diagonal(t)))
        if float(off / tn) <= target_rel:
            break
        bn = block_norms(t, b)
        bn.fill_diagonal_(0.0)
        # threshold: pairs contributing above the target level
        thresh = target_rel * float(tn) / nb
        cand = torch.nonzero(torch.triu(bn, 1) > thresh)
        order = torch.argsort(
            bn[cand[:, 0], cand[:, 1]], descending=True,
        )
        cand = cand[order].tolist()
        used = set()
        rounds_this_sweep = []
        while cand:
            taken = []
            left = []
            used = set()
            for i, j in cand:
                if i in used or j in used:
                    left.append((i, j))
                else:
                    used.add(i)
                    used.add(j)
                    taken.append((i, j))
            cand = left
            rounds_this_sweep.append(taken)
            stats['rounds'] += 1
            stats['pairs'] += len(taken)
            # batched application of the round's (disjoint) rotations
            pr = len(taken)
            idx = torch.stack(
                [
                    torch.cat(
                        [
                            torch.arange(i * b, (i + 1) * b),
                            torch.arange(j * b, (j + 1) * b),
                        ],
                    )
                    for i, j in taken
                ],
            ).to(t.device)
            flat = idx.reshape(-1)
            sub_rows = t.index_select(0, flat).reshape(pr, 2 * b, n)
            subs = torch.gather(
                sub_rows, 2,
                idx.unsqueeze(1).expand(pr, 2 * b, 2 * b),
            )
            _, v = torch.linalg.eigh(
                0.5 * (subs + subs.transpose(-1, -2)),
            )
            # rows, then columns, then accumulate Q
            t.index_copy_(
                0, flat,
                (v.transpose(-1, -2) @ sub_rows).reshape(pr * 2 * b, n),
            )
            sub_cols = (
                t.index_select(1, flat)
                .reshape(n, pr, 2 * b)
                .permute(1, 0, 2)
            )
            t.index_copy_(
                1, flat,
                (sub_cols @ v).permute(1, 0, 2).reshape(n, pr * 2 * b),
            )
            nq = q.size(0)
            q_cols = (
                q.index_select(1, flat)
                .reshape(nq, pr, 2 * b)
                .permute(1, 0, 2)
            )
            q.index_copy_(
                1, flat,
                (q_cols @ v).permute(1, 0, 2).reshape(nq, pr * 2 * b),
            )
    off = torch.linalg.norm(t - torch.diag(torch.diagonal(t)))
    stats['off_final_rel'] = float(off / tn)
    stats['sweeps'] = sweep
    # quality vs direct eigh (true part only; pads never rotate)
    a64 = f_next.to(torch.float64)
    q64 = q[:, :n_true].to(torch.float64)
    d = torch.diagonal(t)[:n_true].to(torch.float64)
    rec = (q64 * d) @ q64.transpose(-1, -2)
    stats['rec_rel'] = float(
        torch.linalg.norm(rec - a64) / torch.linalg.norm(a64),
    )
    eye = torch.eye(n_true, dtype=torch.float64, device=q.device)
    stats['orth'] = float(
        torch.linalg.norm(q64.transpose(-1, -2) @ q64 - eye) / n_true ** 0.5,
    )
    return stats


def time_production_solver(snaps, tags) -> None:
    """Time ops.eigh_batched vs warm_eigh_batched on the CAPTURED real
    factor pairs, grouped by size like the production inverse phase."""
    import time as _time

    from kfac_amd import ops
    from kfac_amd.ops.warm_eigh import warm_eigh_batched

    a_tag, b_tag = tags[1], tags[2]
    groups: dict[int, list[tuple[torch.Tensor, torch.Tensor]]] = {}
    for key in sorted(snaps[a_tag]):
        f0 = snaps[a_tag][key]
        f1 = snaps[b_tag].get(key)
        if f1 is None or f0.shape != f1.shape or f0.size(-1) < 512:
            continue
        groups.setdefault(f0.size(-1), []).append((f0, f1))
    print('\n== production solver timing on captured factors ==')
    for n, items in sorted(groups.items()):
        f0s = torch.stack([a for a, _ in items])
        f1s = torch.stack([b for _, b in items])

        def t(fn, iters=3):
            fn()
            torch.cuda.synchronize()
            t0 = _time.perf_counter()
            for _ in range(iters):
                fn()
            torch.cuda.synchronize()
            return (_time.perf_counter() - t0) / iters * 1000.0

        t_cold = t(lambda: ops.eigh_batched(f1s.clone()))
        _, q0 = ops.eigh_batched(f0s.clone())

        def warm():
            return warm_eigh_batched(f1s, q0, b=32)

        d, q, ok = warm()
        t_warm = t(warm)
        a64 = f1s.to(torch.float64)
        q64 = q.to(torch.float64)
        rec = (q64 * d.to(torch.float64).unsqueeze(1)) @ q64.transpose(-1, -2)
        rec_err = (
            torch.linalg.norm(rec - a64, dim=(-2, -1))
            / torch.linalg.norm(a64, dim=(-2, -1))
        ).max()
        print(
            f'  {len(items)}x{n}: syevd {t_cold:7.1f} ms | '
            f'warm {t_warm:7.1f} ms ok={bool(ok.all())} rec={float(rec_err):.1e} '
            f'speedup {t_cold / t_warm:5.2f}x',
        )


def main() -> None:
    snaps = capture_factors()
    tags = sorted(snaps)
    time_production_solver(snaps, tags)
    for a_tag, b_tag in [(tags[1], tags[2]), (tags[0], tags[1])]:
        print(f'\n== drift {a_tag} -> {b_tag} ==')
        for key in sorted(snaps[a_tag]):
            f0 = snaps[a_tag][key]
            f1 = snaps[b_tag].get(key)
            if f1 is None or f0.shape != f1.shape:
                continue
            n = f0.size(-1)
            if n < 512:
                continue
            rel_change = float(
                torch.linalg.norm(f1 - f0) / torch.linalg.norm(f0),
            )
            for b in (64,):
                st = simulate_block_jacobi(f0, f1, b=b)
                print(
                    f'  {key:<28} n={n:5d} dF={rel_change:.3f} '
                    f'off0={st["off0_rel"]:.3f} sweeps={st["sweeps"]} '
                    f'rounds={st["rounds"]} pairs={st["pairs"]} '
                    f'(dense/sweep={n // b * (n // b - 1) // 2}) '
                    f'off_end={st["off_final_rel"]:.1e} '
                    f'rec={st["rec_rel"]:.1e} orth={st["orth"]:.1e}',
                )


if __name__ == '__main__':
    main()
