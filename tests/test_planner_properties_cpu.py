"""Property-based tests for the planning DPs and the MoE dispatcher.

- cluster_layers is verified OPTIMAL against brute force over all
  contiguous partitions (the reference's layer DP claims min-max
  optimality, layer_construction.py:342).
- The GShard slot assignment never collides, never exceeds capacity,
  and accounts for every kept token exactly once.
"""
import itertools

import torch
from hypothesis import given, settings, strategies as st

from alpa_amd.pipeline_parallel.layer_clustering import cluster_layers


def _brute_min_max(costs, k):
    L = len(costs)
    best = float("inf")
    for cuts in itertools.combinations(range(1, L), k - 1):
        bounds = (0,) + cuts + (L,)
        m = max(sum(costs[a:b]) for a, b in zip(bounds, bounds[1:]))
        best = min(best, m)
    return best


@settings(max_examples=80, deadline=None)
@given(st.lists(st.floats(min_value=0.1, max_value=10.0), min_size=2,
                max_size=9),
       st.integers(min_value=1, max_value=4))
def test_cluster_layers_min_max_optimal(costs, k):
    k = min(k, len(costs))
    ranges = cluster_layers(costs, k)
    assert ranges[0][0] == 0 and ranges[-1][1] == len(costs)
    for (a0, b0), (a1, b1) in zip(ranges, ranges[1:]):
        assert b0 == a1
    got = max(sum(costs[a:b]) for a, b in ranges)
    assert got <= _brute_min_max(costs, k) + 1e-9


@settings(max_examples=50, deadline=None)
@given(st.integers(min_value=1, max_value=64),
       st.sampled_from([2, 4, 8]),
       st.floats(min_value=0.25, max_value=4.0),
       st.integers(min_value=0, max_value=2 ** 31 - 1))
def test_moe_slot_assignment_invariants(n_tokens, n_experts, cap_factor,
                                        seed):
    from alpa_amd.parallel.expert import ExpertParallelMLP, top2_gating
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(n_tokens, n_experts, generator=g)
    C = max(1, int(cap_factor * n_tokens * 2 / n_experts))
    w1g, i1, w2g, i2 = top2_gating(logits, C)
    m = ExpertParallelMLP.__new__(ExpertParallelMLP)
    m.E = n_experts
    zeros = torch.zeros(1, n_experts, dtype=torch.long)
    pos1, keep1, counts = m._assign_slots(i1, C, zeros)
    pos2, keep2, _ = m._assign_slots(i2, C, counts)
    slots = torch.cat([(i1 * C + pos1)[keep1], (i2 * C + pos2)[keep2]])
    # no collisions, all within [0, E*C)
    assert slots.numel() == slots.unique().numel()
    if slots.numel():
        assert int(slots.min()) >= 0
        assert int(slots.max()) < n_experts * C
    # per-expert occupancy never exceeds capacity
    for e in range(n_experts):
        used = int((i1[keep1] == e).sum() + (i2[keep2] == e).sum())
        assert used <= C
    # kept top-1 tokens: exactly the first C per expert in token order
    for e in range(n_experts):
        hits = (i1 == e).nonzero().flatten()
        assert int(keep1[hits].sum()) == min(len(hits), C)


@settings(max_examples=12, deadline=None)
@given(st.sampled_from([2, 4, 8]),
       st.sampled_from([1024, 2560, 4096]),
       st.sampled_from([4096, 16384, 65536]),
       st.booleans())
def test_ilp_plan_wellformed(n, hidden, tokens, force_dp):
    """Any (devices, model size, batch) input yields a well-formed ILP
    plan: a valid mesh factorization, a strategy per node, finite
    objective; force_data_parallel restricts to batch-dim sharding."""
    from alpa_amd.shard_parallel.auto_sharding import solve_gpt_sharding
    plan = solve_gpt_sharding(n, hidden=hidden, layers=2, vocab=2048,
                              tokens=tokens,
                              force_data_parallel=force_dp)
    d0, d1 = plan.mesh_shape
    assert d0 * d1 == n
    assert plan.objective == plan.objective and plan.objective >= 0
    assert plan.choices
    if force_dp:
        assert d1 == 1
        # batch-dim-only sharding: the weight axis of every chosen
        # strategy is None ("colNone"/"rowNone" = replicated weight)
        for name, strat in plan.choices.items():
            assert "col0" not in strat and "col1" not in strat and \
                "row0" not in strat and "row1" not in strat, (name, strat)


def test_training_dp_uses_measured_stage_costs():
    """With a measured stage-cost curve in the DB (the 1-GPU analog of
    the reference's ProfileWorker measurements), the training DP's
    choice follows the MEASURED nonlinearity instead of flops-linear:
    a curve where deep stages are super-linearly expensive pushes the
    optimum toward more, shallower stages."""
    from alpa_amd.mesh_profiling import (CostCurve, MeshProfilingResult,
                                         ProfilingResultDatabase)
    from alpa_amd.pipeline_parallel.stage_construction import \
        training_dp_search
    db = ProfilingResultDatabase()
    db.insert_dummy_mesh_result("mi355x", (1, 1))
    r = db.query("mi355x", (1, 1))
    c = CostCurve()
    # superlinear: 8 layers cost 4x of 4+4 split
    for L, t in ((1.0, 1e-3), (2.0, 2e-3), (4.0, 5e-3), (8.0, 40e-3)):
        c.add(L, t)
    r.op_curves["gpt_stage_cost_hX"] = c
    r.scalars["gpt_stage_cost_hX_batch"] = 8192.0
    flops = [1e12] * 8
    with_meas = training_dp_search(
        4, 8, flops, boundary_act_bytes=1e6,
        layer_param_bytes=[1e8] * 8, db=db,
        stage_cost_curve="gpt_stage_cost_hX", microbatch_tokens=8192.0)
    without = training_dp_search(
        4, 8, flops, boundary_act_bytes=1e6,
        layer_param_bytes=[1e8] * 8, db=db)
    assert with_meas is not None and without is not None
    P_m = with_meas[0]
    # the superlinear measured curve forces multi-stage slicing
    assert P_m >= 2, (with_meas, without)
    # and the reported cost comes from the measured curve's scale
    assert with_meas[3] < 1.0


def test_captured_plan_wellformed_property():
    """Property sweep over random plain MLP chains: every solved plan is
    executable — chosen strategies exist, weight splits divide the
    dims, follow chains carry consistent specs, and the memory
    accounting respects the budget when one is given (the reference's
    plan well-formedness assertions, tests/shard_parallel style)."""
    import torch.nn as nn
    from alpa_amd.shard_parallel import capture_graph, solve_captured

    rng = __import__("random").Random(7)
    for trial in range(6):
        dims = [rng.choice([32, 64, 128])]
        for _ in range(rng.randint(2, 5)):
            dims.append(rng.choice([32, 64, 96, 128, 256]))
        layers = []
        for a, b in zip(dims[:-1], dims[1:]):
            layers += [nn.Linear(a, b), nn.ReLU()]
        torch.manual_seed(trial)
        model = nn.Sequential(*layers)
        x = torch.randn(4, dims[0])
        cap = capture_graph(model, (x,))
        state = sum(12 * p.numel() for p in model.parameters())
        for n, budget in ((2, None), (2, state * 0.8), (4, None)):
            try:
                plan = solve_captured(cap, n, memory_budget=budget,
                                      time_limit=10)
            except AssertionError:
                assert budget is not None  # only budgets may be infeasible
                continue
            dp, tp = plan.mesh_shape
            assert dp * tp == n
            for i, name in plan.choices.items():
                d = cap.ops[i]
                if d.kind != "matmul":
                    continue
                if "_col" in name and not name.endswith("colNone"):
                    assert d.extra["n"] % tp == 0, (name, d.extra)
                if "_row" in name:
                    assert d.extra["k"] % tp == 0, (name, d.extra)
            for i, (ins, out) in plan.specs.items():
                for sp in list(ins) + [out]:
                    for ax in sp:
                        assert ax in (None, 0, 1)
