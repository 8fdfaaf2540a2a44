"""Auto-sharding driver: build the op graph for a model config, run the
ILP over candidate logical mesh shapes, return the best (dp, tp) plan.

Capability analog of the reference's ``run_auto_sharding_pass``
(auto_sharding.py:172) + the logical-shape outer loop: for every
factorization of the device count the ILP scores the whole graph; the
winning mesh shape + per-matmul strategy becomes the executed plan
(tensor-parallel layer construction in parallel/layers.py).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

from ..global_env import global_config
from .ir import Graph, MeshModel, Node
from .solver import SolveResult, solve
from .strategies import (elemwise_follow_node, embedding_strategies,
                         loss_strategies, matmul_strategies)


def build_gpt_graph(mesh: MeshModel, hidden: int, layers: int, vocab: int,
                    tokens: int, ffn_mult: int = 4,
                    dtype_bytes: int = 2) -> Graph:
    """GPT train-step op graph: embed -> L x (qkv, attn, out, fc1, gelu,
    fc2) -> ln_f -> lm_head -> loss.  LayerNorms/attention/residuals are
    follow nodes (elementwise-aligned)."""
    g = Graph()
    embed = g.add(Node("wte", "embedding", [],
                       out_bytes=float(tokens) * hidden * dtype_bytes,
                       strategies=embedding_strategies(mesh, tokens, vocab,
                                                       hidden, dtype_bytes)))
    prev = embed
    h_bytes = float(tokens) * hidden * dtype_bytes
    for l in range(layers):
        ln1 = g.add(elemwise_follow_node(f"b{l}.ln1", prev, h_bytes))
        qkv = g.add(Node(
            f"b{l}.qkv", "matmul", [ln1],
            out_bytes=3.0 * tokens * hidden * dtype_bytes,
            strategies=matmul_strategies(mesh, tokens, hidden, 3 * hidden,
                                         dtype_bytes)))
        attn = g.add(elemwise_follow_node(f"b{l}.attn", qkv, h_bytes))
        out = g.add(Node(
            f"b{l}.out", "matmul", [attn],
            out_bytes=h_bytes,
            strategies=matmul_strategies(mesh, tokens, hidden, hidden,
                                         dtype_bytes)))
        res1 = g.add(elemwise_follow_node(f"b{l}.res1", out, h_bytes))
        ln2 = g.add(elemwise_follow_node(f"b{l}.ln2", res1, h_bytes))
        fc1 = g.add(Node(
            f"b{l}.fc1", "matmul", [ln2],
            out_bytes=float(tokens) * ffn_mult * hidden * dtype_bytes,
            strategies=matmul_strategies(mesh, tokens, hidden,
                                         ffn_mult * hidden, dtype_bytes)))
        gelu = g.add(elemwise_follow_node(
            f"b{l}.gelu", fc1, float(tokens) * ffn_mult * hidden *
            dtype_bytes))
        fc2 = g.add(Node(
            f"b{l}.fc2", "matmul", [gelu],
            out_bytes=h_bytes,
            strategies=matmul_strategies(mesh, tokens, ffn_mult * hidden,
                                         hidden, dtype_bytes)))
        prev = g.add(elemwise_follow_node(f"b{l}.res2", fc2, h_bytes))
    lnf = g.add(elemwise_follow_node("ln_f", prev, h_bytes))
    head = g.add(Node(
        "lm_head", "matmul", [lnf],
        out_bytes=float(tokens) * vocab * dtype_bytes,
        strategies=matmul_strategies(mesh, tokens, hidden, vocab,
                                     dtype_bytes)))
    g.add(Node("loss", "loss", [head], out_bytes=0.0,
               strategies=loss_strategies(mesh, tokens, vocab, dtype_bytes)))
    return g


def build_mlp_graph(mesh: MeshModel, hidden: int, num_layers: int,
                    tokens: int, dtype_bytes: int = 4) -> Graph:
    """The 4-layer MLP of the plumbing benchmark / solver tests (analog of
    playground/auto_sharding_solver/test_solver_mlp.py)."""
    from .ir import Strategy
    g = Graph()
    in_bytes = float(tokens) * hidden * dtype_bytes
    input_strats = [
        Strategy(name=f"b{ax}", in_specs=[], out_spec=(ax, None),
                 compute_cost=0.0, comm_cost=0.0,
                 memory=in_bytes / mesh.axis_size(ax))
        for ax in (None, 0, 1)
    ]
    prev = g.add(Node("input", "input", [], out_bytes=in_bytes,
                      strategies=input_strats))
    for l in range(num_layers):
        n_out = hidden * 4 if l % 2 == 0 else hidden
        n_in = hidden if l % 2 == 0 else hidden * 4
        node = g.add(Node(
            f"l{l}", "matmul", [prev],
            out_bytes=float(tokens) * n_out * dtype_bytes,
            strategies=matmul_strategies(mesh, tokens, n_in, n_out,
                                         dtype_bytes)))
        prev = node
    return g


@dataclass
class ShardingPlan:
    mesh_shape: Tuple[int, int]
    objective: float
    choices: Dict[str, str]  # node name -> strategy name


def plan_to_logical_shape(plan: "ShardingPlan") -> Tuple[int, int]:
    """Map the solver's (mesh shape, strategy names) to the executor's
    (dp, tp) convention: axis 0 = batch/data, axis 1 = tensor/model.

    Permutes solver axes if it used axis 1 for the batch; replicated
    everywhere degenerates to pure DP.
    """
    d0, d1 = plan.mesh_shape
    n = d0 * d1
    batch_use = {0: 0, 1: 0}
    weight_use = {0: 0, 1: 0}
    for name, strat in plan.choices.items():
        if "_" not in strat:
            continue
        b_tok, w_tok = strat.split("_", 1)
        b_ax = b_tok[1:]
        if b_ax in ("0", "1"):
            batch_use[int(b_ax)] += 1
        for prefix in ("col", "row", "vocab", "v"):
            if w_tok.startswith(prefix):
                ax = w_tok[len(prefix):]
                if ax in ("0", "1"):
                    weight_use[int(ax)] += 1
    dp_ax = max(batch_use, key=batch_use.get) if any(batch_use.values()) \
        else None
    tp_ax = max(weight_use, key=weight_use.get) if any(weight_use.values()) \
        else None
    if dp_ax is not None and tp_ax == dp_ax:
        tp_ax = 1 - dp_ax
    dp = plan.mesh_shape[dp_ax] if dp_ax is not None else 1
    tp = plan.mesh_shape[tp_ax] if tp_ax is not None else 1
    if dp * tp != n:
        # uncovered axis (fully replicated direction): absorb into dp
        dp = n // tp
    return (dp, tp)


def solve_gpt_sharding(num_devices: int, hidden: int, layers: int,
                       vocab: int, tokens: int,
                       memory_budget: Optional[float] = None,
                       force_data_parallel: bool = False,
                       time_limit: Optional[float] = None) -> ShardingPlan:
    """Outer loop over logical mesh shapes + inner ILP (reference
    compile_shard_executable tries logical shapes; ILP scores each)."""
    from .mesh_search import factorizations
    if time_limit is None:
        # per-mesh-shape slice of the configured total (reference uses a
        # 600 s PuLP limit, auto_sharding.py:829-836)
        time_limit = min(30.0, global_config.solver_timeout / 8)
    best: Optional[ShardingPlan] = None
    for (d0, d1) in factorizations(num_devices):
        if force_data_parallel and d1 != 1:
            continue
        mesh = MeshModel((d0, d1), alpha=global_config.mesh_alpha,
                         beta=global_config.mesh_beta)
        g = build_gpt_graph(mesh, hidden, layers, vocab, tokens)
        if force_data_parallel:
            # restrict the ILP to batch-dim sharding (reference
            # auto_sharding.py:239-245)
            for node in g.nodes:
                if node.follow is not None or not node.strategies:
                    continue
                kept = [st for st in node.strategies
                        if st.out_spec[1] is None and
                        st.out_spec[0] is not None and
                        all(sp[1] is None for sp in st.in_specs)]
                if kept:
                    node.strategies = kept
        res = solve(g, mesh, memory_budget=memory_budget,
                    time_limit=time_limit)
        if not res.feasible:
            continue
        plan = ShardingPlan(
            mesh_shape=(d0, d1), objective=res.objective,
            choices={g.nodes[i].name: g.nodes[i].strategies[r].name
                     for i, r in res.choices.items()
                     if g.nodes[i].follow is None})
        if best is None or plan.objective < best.objective:
            best = plan
    assert best is not None, "no feasible sharding found"
    if global_config.print_strategy:
        print(f"[auto_sharding] mesh {best.mesh_shape} "
              f"objective {best.objective:.4f}")
        for k, v in list(best.choices.items())[:12]:
            print(f"  {k}: {v}")
    return best


# ----------------------------------------------------------------------
# Captured graphs: arbitrary user programs
# ----------------------------------------------------------------------


def build_captured_graph(cap, mesh: MeshModel, dtype_bytes: int = 2,
                         train: bool = True,
                         fixed_b: Optional[int] = None) -> Graph:
    """Lower a capture.CapturedGraph (torch.fx trace of the USER's
    program) to the ILP op graph: per-node strategies enumerated for the
    traced shapes — the analog of the reference's per-instruction
    strategy construction over real HLO
    (playground/auto_sharding_solver/hlo.py:664-830), replacing the
    model_hint template path.

    ``fixed_b`` restricts every node's batch split to that mesh axis
    (None = no batch split): the executor feeds per-rank batch shards,
    so the batch axis must be uniform across the program.  Per-node
    WEIGHT axes stay free — mixed plans (shard the MLP, replicate the
    attention) remain expressible.
    """
    import math as _math
    from .strategies import (input_strategies,
                             replicated_feature_strategies)
    g = Graph()
    for d in cap.ops:
        nelem = float(_math.prod(d.out_shape)) if d.out_shape else 0.0
        out_bytes = nelem * dtype_bytes
        if d.kind == "input":
            strats = input_strategies(mesh, out_bytes)
            g.add(Node(d.name, "input", [], out_bytes, strategies=strats))
        elif d.kind == "matmul" or d.kind == "conv":
            tokens = int(nelem // max(d.extra["n"], 1))
            strats = matmul_strategies(mesh, tokens, d.extra["k"],
                                       d.extra["n"], dtype_bytes,
                                       train=train)
            # only divisible shards are executable by the parallel layers
            # (conv row-sharding splits IN CHANNELS, not k=cin*kh*kw;
            # grouped convs stay replicated)
            def _divisible(st):
                if d.kind == "conv" and d.extra.get("groups", 1) != 1:
                    return st.out_spec[1] is None and (
                        not st.in_specs or st.in_specs[0][1] is None)
                w_ax = st.out_spec[1]
                if w_ax is not None and \
                        d.extra["n"] % mesh.shape[w_ax] != 0:
                    return False
                in_ax = st.in_specs[0][1] if st.in_specs else None
                in_div = d.extra.get("cin", d.extra["k"])
                if in_ax is not None and \
                        in_div % mesh.shape[in_ax] != 0:
                    return False
                return True
            strats = [st for st in strats if _divisible(st)]
            g.add(Node(d.name, "matmul", list(d.inputs), out_bytes,
                       strategies=strats))
        elif d.kind == "embedding":
            tokens = int(nelem // max(d.extra["h"], 1))
            strats = embedding_strategies(mesh, tokens, d.extra["vocab"],
                                          d.extra["h"], dtype_bytes)
            strats = [st for st in strats
                      if "vocab" not in st.name or
                      d.extra["vocab"] % mesh.shape[
                          int(st.name.split("vocab")[1])] == 0]
            g.add(Node(d.name, "embedding", [], out_bytes,
                       strategies=strats))
        elif d.kind == "elemwise" and d.inputs:
            g.add(elemwise_follow_node(d.name, d.inputs[0], out_bytes))
        elif d.kind in ("norm", "opaque", "output") or not d.inputs:
            strats = replicated_feature_strategies(
                mesh, out_bytes, n_inputs=max(len(d.inputs), 1),
                name=d.kind)
            g.add(Node(d.name, d.kind, list(d.inputs), out_bytes,
                       strategies=strats))
        else:
            g.add(elemwise_follow_node(d.name, d.inputs[0], out_bytes))
    # uniform batch axis across the program (the executor feeds fixed
    # per-rank batch shards); weight axes stay per-node free
    for node in g.nodes:
        if node.follow is not None or not node.strategies:
            continue
        kept = [st for st in node.strategies
                if st.out_spec[0] == fixed_b]
        if kept:
            node.strategies = kept
    return g


@dataclass
class CapturedPlan:
    """Solved sharding for a captured program: logical mesh + one
    strategy per op (follow nodes resolved), ready for
    plan_apply.apply_captured_plan.  JSON-serializable for the
    solve-once/replay workflow (reference ParallelPlan +
    LoadSolutionParallelArgs, benchmark_parallel_utils.py:39)."""
    mesh_shape: Tuple[int, int]            # (dp, tp): axis 0 batch, 1 tp
    objective: float
    #: op index -> strategy name (owners only)
    choices: Dict[int, str]
    #: op index -> (in_specs, out_spec) of the chosen strategy
    specs: Dict[int, Tuple[tuple, tuple]]

    def save(self, path: str):
        import json
        d = {"mesh_shape": list(self.mesh_shape),
             "objective": self.objective,
             "choices": {str(i): n for i, n in self.choices.items()},
             "specs": {str(i): [[list(sp) for sp in ins], list(out)]
                       for i, (ins, out) in self.specs.items()}}
        with open(path, "w") as f:
            json.dump(d, f, indent=2)

    @staticmethod
    def load(path: str) -> "CapturedPlan":
        import json
        with open(path) as f:
            d = json.load(f)

        def tup(x):
            return tuple(None if v is None else int(v) for v in x)

        return CapturedPlan(
            mesh_shape=tuple(d["mesh_shape"]),
            objective=d["objective"],
            choices={int(i): n for i, n in d["choices"].items()},
            specs={int(i): ([tup(sp) for sp in ins], tup(out))
                   for i, (ins, out) in d["specs"].items()})


def solve_captured(cap, num_devices: int,
                   memory_budget: Optional[float] = None,
                   force_data_parallel: bool = False,
                   time_limit: Optional[float] = None,
                   dtype_bytes: int = 2,
                   train: bool = True,
                   mesh_shape: Optional[Tuple[int, int]] = None
                   ) -> CapturedPlan:
    """Outer loop over (dp, tp) factorizations x inner ILP over the
    captured graph (reference compile_shard_executable logical-shape
    loop + _call_solver_serialized_args)."""
    from .mesh_search import factorizations
    if time_limit is None:
        time_limit = min(30.0, global_config.solver_timeout / 8)
    best = None
    for (dp, tp) in factorizations(num_devices):
        if force_data_parallel and tp != 1:
            continue
        if mesh_shape is not None and (dp, tp) != tuple(mesh_shape):
            continue
        mesh = MeshModel((dp, tp), alpha=global_config.mesh_alpha,
                         beta=global_config.mesh_beta)
        fixed_b = 0 if dp > 1 else None
        g = build_captured_graph(cap, mesh, dtype_bytes, train, fixed_b)
        res = solve(g, mesh, memory_budget=memory_budget,
                    time_limit=time_limit,
                    edge_cost_factor=2.0 if train else 1.0)
        if not res.feasible:
            continue
        choices, specs = {}, {}
        for i, r in res.choices.items():
            node = g.nodes[i]
            if node.follow is not None:
                # resolve to owner's strategy for spec bookkeeping
                owner = i
                seen = set()
                while g.nodes[owner].follow is not None and \
                        owner not in seen:
                    seen.add(owner)
                    owner = g.nodes[owner].follow
                st = g.nodes[owner].strategies[res.choices[owner]]
                specs[i] = ([st.out_spec] * max(len(node.inputs), 1),
                            st.out_spec)
                continue
            st = node.strategies[r]
            choices[i] = st.name
            specs[i] = (list(st.in_specs), st.out_spec)
        plan = CapturedPlan((dp, tp), res.objective, choices, specs)
        if best is None or plan.objective < best.objective:
            best = plan
    assert best is not None, "no feasible sharding found for captured graph"
    if global_config.print_strategy:
        print(f"[auto_sharding/captured] mesh {best.mesh_shape} "
              f"objective {best.objective:.4f}")
        for i, name in list(best.choices.items())[:16]:
            print(f"  {cap.ops[i].name}: {name}")
    return best
