"""ACOAgent: ChebConv actor + differentiable queueing critic + semi-analytic
policy gradient.

Clean-room reimplementation of ``gnn_offloading_agent.py:64-453`` on torch:

  * ``forward``       — GNN λ prediction → contention fixed point → link/node
                        delays → N×N delay matrix (autograd graph kept);
                        reference ``:211-276``.
  * ``forward_env``   — forward + APSP + offloading + empirical evaluation;
                        reference ``:278-291``.
  * ``forward_backward`` — the full semi-analytic policy gradient: critic
                        loss over the routes matrix, grad wrt routes via
                        autograd, the route-bias suffix-sum VJP ("Method 2",
                        reference ``:384-416``) computed in closed form as a
                        per-route prefix scan, the 0.001·MSE anchor
                        (``:440-444``), and the actor VJP with the N×N
                        cotangent (``:448``); gradient sets are memorised for
                        replay.
  * ``replay``        — sample stored gradient sets, apply Adam (Keras
                        semantics: eps 1e-7, per-tensor clipnorm 1.0,
                        max_norm constraints after each step), ε decay;
                        reference ``:156-169``.

Under the canonical extended-edge ordering (graphs.py) the reference's
``maps_ol_el`` is ``arange(E)`` and the per-extended-edge unit-delay vector
is literally ``cat([link_delay, node_delay])`` — no scatter needed.
"""

from __future__ import annotations

import dataclasses
import random
from collections import deque
from typing import List, Optional

import numpy as np
import torch

from .env import AdhocCloudEnv, apsp
from .graphs import CaseGraph, JobInstance
from .models.chebconv import ChebConvStack
from .queueing import ConflictCSR, actor_delays, delay_matrix, delay_with_fallback, fixed_point_mu
from .utils import checkpoint as ckpt_io


@dataclasses.dataclass
class AgentConfig:
    """Reference flag names and defaults (``gnn_offloading_agent.py:42-60``)
    plus the MI355X-native knobs (K, device, dtype, support mode)."""
    T: int = 1000
    prob: bool = False
    learning_rate: float = 1e-4
    learning_decay: float = 1.0
    arrival_scale: float = 0.1
    epochs: int = 201
    num_layer: int = 5
    dropout: float = 0.0
    weight_decay: float = 5e-4
    epsilon: float = 1.0
    epsilon_min: float = 0.001
    epsilon_decay: float = 0.985
    gamma: float = 1.0
    batch: int = 100
    # new knobs
    K: int = 2
    hidden: int = 32
    device: str = "cpu"
    dtype: str = "float64"          # oracle fp64 on CPU; fp32 on GPU
    fp_iters: int = 10
    seed: Optional[int] = None


class ACOAgent:
    """Single-case actor-critic offloading agent with the reference's API
    and training dynamics (see module docstring); the batched device path
    is ``engine.EpisodeEngine``."""

    def __init__(self, cfg: AgentConfig, memory_size: int = 5000):
        self.cfg = cfg
        self.dtype = getattr(torch, cfg.dtype)
        self.device = torch.device(cfg.device)
        self.model = ChebConvStack(
            in_dim=4, hidden=cfg.hidden, out_dim=1, num_layer=cfg.num_layer,
            K=cfg.K, dropout=cfg.dropout, dtype=self.dtype, seed=cfg.seed,
        ).to(self.device)
        self.optimizer = torch.optim.Adam(
            self.model.parameters(), lr=cfg.learning_rate, eps=1e-7)
        self.memory = deque(maxlen=memory_size)
        self.epsilon = cfg.epsilon
        self._step_count = 0
        self._support_cache = {}

    # -- checkpointing (reference :125-132) -----------------------------------
    def load(self, directory: str):
        latest = ckpt_io.latest_checkpoint(directory)
        if latest:
            ckpt_io.load(self.model, latest)
            print("Actor loaded " + latest)

    def save(self, ckpt_path: str):
        ckpt_io.save(self.model, ckpt_path)

    # -- support / tensors -----------------------------------------------------
    def _support(self, g: CaseGraph) -> ConflictCSR:
        key = id(g)
        got = self._support_cache.get(key)
        if got is None or got[0] is not g:
            ext = g.ext
            got = (g, ConflictCSR(ext.ext_indptr, ext.ext_indices,
                                  device=self.device),
                   ConflictCSR(g.conf_indptr, g.conf_indices,
                               device=self.device))
            self._support_cache = {key: got}   # keep one case hot
        return got[1], got[2]

    def _t(self, a, dtype=None):
        return torch.as_tensor(a, dtype=dtype or self.dtype, device=self.device)

    # -- actor forward (reference :211-276) ------------------------------------
    def forward(self, g: CaseGraph, jobs: JobInstance):
        ext = g.ext
        support, conf = self._support(g)
        x = self._t(ext.features(jobs))
        lam = self.model(x, support)[:, 0]                  # (Ē,)
        E = g.num_links
        lam_link = lam[:E]
        lam_node = lam[E:]                                  # canonical order
        link_delay, node_delay = actor_delays(
            lam_link, lam_node, self._t(g.link_rates), self._t(g.cf_degs),
            conf, self._t(g.proc_bws[ext.comp_nodes]), g.T, self.cfg.fp_iters)
        dm = delay_matrix(link_delay, node_delay,
                          torch.as_tensor(g.edges, device=self.device),
                          torch.as_tensor(ext.comp_nodes, device=self.device),
                          g.num_nodes)
        return dm, link_delay, node_delay

    def forward_gcn(self, g: CaseGraph, jobs: JobInstance):
        """3-feature variant (reference ``forward_gcn``,
        gnn_offloading_agent.py:171-209): the GNN output is used directly as
        per-link/per-node delays (no queueing fixed point).  Unused by the
        harnesses; kept for API parity."""
        ext = g.ext
        support, _ = self._support(g)
        x4 = ext.features(jobs)
        x = self._t(x4[:, :3])                  # [loop, rate, job]
        if self.model.layers[0].weight.shape[1] != 3:
            raise ValueError("forward_gcn needs a model with in_dim=3")
        lam = self.model(x, support)[:, 0]
        E = g.num_links
        dm = delay_matrix(lam[:E], lam[E:],
                          torch.as_tensor(g.edges, device=self.device),
                          torch.as_tensor(ext.comp_nodes, device=self.device),
                          g.num_nodes)
        return dm

    # -- reference state/predict API (gnn_offloading_agent.py:134-154) ----------
    def makestate(self, g: CaseGraph, jobs: JobInstance):
        """Bundle the GNN inputs like the reference's ``makestate``; the
        support entry is the case itself (its CSR operator pair is cached
        per case by ``_support``)."""
        return {"node_features": g.ext.features(jobs), "support": g}

    def predict(self, state):
        """Run the actor GNN on a ``makestate`` bundle (reference
        ``predict``/``act``): returns the raw (Ē, 1) head output."""
        g = state["support"]
        support, _ = self._support(g)
        x = self._t(np.asarray(state["node_features"]))
        return self.model(x, support)

    def act(self, state):
        return self.predict(state)

    # -- forward + environment (reference :278-291) -----------------------------
    def _env_step(self, env: AdhocCloudEnv, explore: float,
                  rng: Optional[np.random.RandomState] = None):
        g = env.g
        dm, link_delay, node_delay = self.forward(g, env.jobs)
        link_delay_np = link_delay.detach().cpu().numpy()
        dm_np = dm.detach().cpu().numpy()
        sp_gnn = apsp(g, link_delay_np)
        np.fill_diagonal(sp_gnn, np.diagonal(dm_np))
        decisions, delay_est = env.offloading(sp_gnn, g.sp_hop, explore,
                                              self.cfg.prob, rng)
        delay_links, delay_nodes, delay_unit = env.run()
        return dm, dm_np, delay_links, delay_nodes, delay_unit

    def forward_env(self, env: AdhocCloudEnv,
                    rng: Optional[np.random.RandomState] = None):
        with torch.no_grad():
            _, _, delay_links, delay_nodes, delay_unit = self._env_step(
                env, 0.0, rng)
        return delay_links, delay_nodes, delay_unit

    # -- critic + backward (reference :293-453) ---------------------------------
    def forward_backward(self, env: AdhocCloudEnv, explore: float = 0.0,
                         rng: Optional[np.random.RandomState] = None):
        g, jobs = env.g, env.jobs
        ext = g.ext
        E, Ee, J = g.num_links, ext.num_edges_ext, jobs.num_jobs

        dm, dm_np, delay_links, delay_nodes, delay_unit = self._env_step(
            env, explore, rng)

        # routes matrix Ē×J + per-job extended-edge sequences (forward order,
        # self-loop last) — reference :310-331
        route_seqs: List[np.ndarray] = []
        routes_np = np.zeros((Ee, J))
        for j, flow in enumerate(env.flows):
            links = env.route_links(flow)                   # real link ids
            seq = np.concatenate([links, [ext.node_vedge[flow.dst]]])
            route_seqs.append(seq)
            routes_np[seq, j] = 1.0

        # ---- critic: loss over routes, grad wrt routes (reference :333-374)
        routes = self._t(routes_np).requires_grad_(True)
        jobs_load = self._t((jobs.rates * jobs.ul)[:, None])        # (J,1)
        jobs_data = self._t((jobs.ul + jobs.dl)[None, :])           # (1,J)
        _, conf = self._support(g)
        link_load = (routes @ jobs_load)[:, 0]                      # (Ē,)
        lam_link, lam_node = link_load[:E], link_load[E:]
        mu = fixed_point_mu(lam_link, self._t(g.link_rates),
                            self._t(g.cf_degs), conf, self.cfg.fp_iters)
        link_d = delay_with_fallback(lam_link, mu, g.T, 101.0)
        node_d = delay_with_fallback(lam_node,
                                     self._t(g.proc_bws[ext.comp_nodes]),
                                     g.T, 100.0)
        unit_edge = torch.cat([link_d, node_d])                     # (Ē,)
        delay_job_edge = torch.maximum(jobs_data * unit_edge[:, None] * routes,
                                       routes)
        loss_fn = delay_job_edge.sum()
        (grad_routes,) = torch.autograd.grad(loss_fn, routes)
        grad_routes_np = grad_routes.detach().cpu().numpy()

        # ---- route-bias VJP ("Method 2", reference :384-416), closed form:
        # bias[e_k, j] = sum of unit delays from e_k to the destination
        # self-loop, so d(sum_j -grad_routes·bias)/d(unit[e]) accumulates the
        # *forward prefix sum* of -grad_routes along each route.
        grad_edge = np.zeros(Ee)
        for j, seq in enumerate(route_seqs):
            np.add.at(grad_edge, seq, -np.cumsum(grad_routes_np[seq, j]))

        # scatter to the N×N cotangent (assignment per extended edge,
        # reference :410-416)
        grad_dist = np.zeros((g.num_nodes, g.num_nodes))
        e0, e1 = g.edges[:, 0], g.edges[:, 1]
        grad_dist[e0, e1] = grad_edge[:E]
        grad_dist[e1, e0] = grad_edge[:E]
        grad_dist[ext.comp_nodes, ext.comp_nodes] = grad_edge[E:]

        # ---- MSE anchor (reference :440-444)
        du = delay_unit.copy()
        du[np.isinf(du)] = np.nan
        with np.errstate(invalid="ignore"):
            diff = dm_np - du
        sq = diff ** 2
        valid = ~np.isnan(sq)
        # nanmean semantics without the all-NaN RuntimeWarning (zero-job
        # instances have an empty unit matrix)
        loss_mse = float(sq[valid].mean()) if valid.any() else float("nan")
        grad_dist += np.nan_to_num(0.001 * diff, nan=0.0)

        # ---- actor VJP (reference :448)
        params = list(self.model.parameters())
        grads = torch.autograd.grad(
            dm, params, grad_outputs=self._t(grad_dist), allow_unused=True)
        grads = [torch.zeros_like(p) if gr is None else gr.detach()
                 for p, gr in zip(params, grads)]
        self.memorize(grads, float(loss_fn.detach()), loss_mse)

        return dm_np, delay_links, delay_nodes, delay_unit, list(env.flows), \
            float(loss_fn.detach()), loss_mse

    # -- scalar logging (reference log_init/log_scalar, :455-469) ---------------
    def log_init(self, logdir="logs"):
        import datetime
        import os
        stamp = datetime.datetime.now().strftime("%Y%m%d-%H%M%S")
        base = os.path.join(logdir, "gradient_tape", stamp)
        self.train_summary_writer = _ScalarLogger(
            os.path.join(base, "train.jsonl"))
        self.test_summary_writer = _ScalarLogger(
            os.path.join(base, "test.jsonl"))

    def log_scalar(self, name, variable, step, test=False):
        w = self.test_summary_writer if test else self.train_summary_writer
        w.scalar(name, variable, step)

    # -- replay memory (reference :141-169) -------------------------------------
    def memorize(self, grads, loss, reward):
        self.memory.append((grads, loss, reward))

    def _apply_one(self, grads):
        cfg = self.cfg
        if cfg.learning_decay != 1.0:
            lr = cfg.learning_rate * cfg.learning_decay ** (self._step_count / 100.0)
            for group in self.optimizer.param_groups:
                group["lr"] = lr
        for p, gr in zip(self.model.parameters(), grads):
            n = gr.norm()
            p.grad = gr if n <= 1.0 else gr * (1.0 / n)     # Keras clipnorm
        self.optimizer.step()
        self.model.apply_constraints()
        self._step_count += 1

    def replay(self, batch_size: int) -> float:
        if len(self.memory) < batch_size:
            return float("nan")
        minibatch = random.sample(list(self.memory), batch_size)
        losses = []
        for grads, loss, _ in minibatch:
            self._apply_one(grads)
            losses.append(loss)
        if self.epsilon > self.cfg.epsilon_min:
            self.epsilon *= self.cfg.epsilon_decay
        return float(np.nanmean(losses))


class _ScalarLogger:
    """JSONL scalar logger — the MI355X-native stand-in for the reference's
    TF summary writers (``gnn_offloading_agent.py:455-469``; commented out in
    its harnesses).  One line per scalar: {"name", "value", "step"}."""

    def __init__(self, path: str):
        import os
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        self._fh = open(path, "a", buffering=1)

    def scalar(self, name: str, value, step: int):
        import json
        self._fh.write(json.dumps(
            {"name": name, "value": float(value), "step": int(step)}) + "\n")

    def close(self):
        self._fh.close()
