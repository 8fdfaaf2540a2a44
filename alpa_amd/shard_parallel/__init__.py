"""Intra-operator auto-sharding: graph capture -> ILP -> plan execution
(reference alpa/shard_parallel/)."""
from .auto_sharding import (CapturedPlan, ShardingPlan, build_captured_graph,
                            solve_captured, solve_gpt_sharding)
from .capture import CapturedGraph, OpDesc, capture_graph
from .manual_sharding import apply_manual_sharding
from .plan_apply import apply_captured_plan, auto_shard
