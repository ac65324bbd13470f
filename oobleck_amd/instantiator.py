# Pipeline-instantiation planning helpers — dependency-free restatements of
# the reference's pyomo/glpk-based pieces (pyomo is not installed in this
# environment; SURVEY.md §2 "Pipeline instantiator"):
#
#  * enumerate_instantiation_options: the knapsack DP of
#    /root/reference/oobleck/planning/instantiator.py:224-252.
#  * distribute_batch: the MINLP of instantiator.py:254-329 — minimize
#    sum_i (T_i/s_i * nb_i - avg)^2  s.t.  sum_i nb_i * x_i = B, nb_i >= 1
#    integer — solved exactly by bounded enumeration (the instance counts
#    and B are tiny: B = global/microbatch <= a few hundred, <= ~5
#    heterogeneous templates).
#
#  * partition_microbatches: the OobleckSampler partitioning arithmetic
#    (/root/reference/oobleck/execution/dataloader.py:43-100): each pipeline
#    consumes its own num_microbatches share of the global batch per step.
from __future__ import annotations

from collections import defaultdict
from dataclasses import dataclass


@dataclass(frozen=True)
class TemplateSpec:
    """The attributes the planning arithmetic reads off a PipelineTemplate
    (pipeline_template.pyi surface: _num_nodes, _num_gpus_per_node,
    _iteration_time, get_stages())."""
    num_nodes: int
    num_gpus_per_node: int
    iteration_time: float
    num_stages: int


def enumerate_instantiation_options(templates: list[TemplateSpec],
                                    num_nodes: int) -> list[dict[TemplateSpec, int]]:
    dp: list[list[list[dict[TemplateSpec, int]]]] = [
        [[] for _ in range(num_nodes + 1)] for _ in range(len(templates) + 1)]
    for i in range(1, len(templates) + 1):
        dp[i][0] = [defaultdict(int)]
        for j in range(1, num_nodes + 1):
            dp[i][j] = [combo.copy() for combo in dp[i - 1][j]]
            if templates[i - 1].num_nodes <= j:
                for combo in dp[i][j - templates[i - 1].num_nodes]:
                    new_combo = combo.copy()
                    new_combo[templates[i - 1]] += 1
                    dp[i][j].append(new_combo)
    return dp[-1][-1]


def distribute_batch(global_num_microbatch: int,
                     num_instances_set: dict[TemplateSpec, int]) -> dict[TemplateSpec, int] | None:
    """Exact integer minimizer of sum((T_i/s_i*nb_i - avg)^2) subject to
    sum(nb_i * x_i) = B, nb_i >= 1.  Same objective/constraints as the
    reference's pyomo model; exhaustive over the (tiny) feasible set with
    a safety cap, falling back to a rounding + local-search heuristic."""
    specs = list(num_instances_set.keys())
    x = [num_instances_set[t] for t in specs]
    w = [t.iteration_time / t.num_stages for t in specs]
    n = len(specs)
    B = global_num_microbatch
    if n == 0 or B < sum(x):
        return None

    def cost(nb: list[int]) -> float:
        vals = [w[i] * nb[i] for i in range(n)]
        avg = sum(vals) / n
        return sum((v - avg) ** 2 for v in vals)

    best: tuple[float, list[int]] | None = None
    budget = [10_000_000]  # enumeration cap

    def rec(i: int, remaining: int, cur: list[int]):
        nonlocal best
        if budget[0] <= 0:
            return
        if i == n - 1:
            if remaining >= x[i] and remaining % x[i] == 0:
                nb = cur + [remaining // x[i]]
                c = cost(nb)
                if best is None or c < best[0]:
                    best = (c, nb)
            budget[0] -= 1
            return
        max_nb = (remaining - sum(x[i + 1:])) // x[i]
        for nb_i in range(1, max_nb + 1):
            rec(i + 1, remaining - nb_i * x[i], cur + [nb_i])
            if budget[0] <= 0:
                return

    rec(0, B, [])
    if best is None:
        return None
    return {t: nb for t, nb in zip(specs, best[1])}


def partition_microbatches(num_instances_set: dict[TemplateSpec, int],
                           nb_per_template: dict[TemplateSpec, int]) -> list[int]:
    """Flatten to a per-pipeline microbatch count list, template order
    preserved (dataloader.py:43-100: the sampler walks pipelines in plan
    order; pipeline i consumes num_microbatches[i] microbatches per step)."""
    out: list[int] = []
    for t, count in num_instances_set.items():
        out.extend([nb_per_template[t]] * count)
    return out
