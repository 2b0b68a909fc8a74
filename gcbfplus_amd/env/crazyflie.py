"""CrazyFlie env: full 12-state quadrotor with an internal low-level LQR
velocity controller (reference ``gcbfplus/env/crazyflie.py``).

State (x, y, z, psi, theta, phi, u, v, w, r, q, p) — body-frame velocities
uvw and angular rates rqp; action = 4 world-frame velocity targets
(vx, vy, vz, r), scaled by (2, 2, 0.5, 0.1) and tracked by an LQR on the
9-dim low-level state (:305-351, 423-486). Integration: RK4 of the full
rigid-body dynamics with the LL controller re-evaluated per stage (:620-625).
Edge features: 12-dim world-frame [rel pos, rel vel, rel body-z, rel omega]
(:223-245). LQR gains from scipy's continuous ARE (replacing the reference's
python-`control` ct.lqr, :488-536).
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import numpy as np
import torch
from torch import Tensor

from ..utils.graph import GraphBatch
from .base import StepResult
from .linear_drone import LinearDrone
from .obstacle import Sphere
from .utils import sample_starts_goals

# state indices (reference :58)
X, Y, Z, PSI, THETA, PHI, U, V, W, R_, Q_, P_ = range(12)


def rotmat(phi: Tensor, theta: Tensor, psi: Tensor) -> Tensor:
    """(...,) angles -> (..., 3, 3) body->world rotation (reference :19-33)."""
    c_phi, s_phi = torch.cos(phi), torch.sin(phi)
    c_th, s_th = torch.cos(theta), torch.sin(theta)
    c_psi, s_psi = torch.cos(psi), torch.sin(psi)
    rows = [
        torch.stack([c_psi * c_th, c_psi * s_th * s_phi - s_psi * c_phi,
                     c_psi * s_th * c_phi + s_psi * s_phi], dim=-1),
        torch.stack([s_psi * c_th, s_psi * s_th * s_phi + c_psi * c_phi,
                     s_psi * s_th * c_phi - c_psi * s_phi], dim=-1),
        torch.stack([-s_th, c_th * s_phi, c_th * c_phi], dim=-1),
    ]
    return torch.stack(rows, dim=-2)


class CrazyFlie(LinearDrone):
    PARAMS = {
        "drone_radius": 0.05,
        "comm_radius": 1.0,
        "n_rays": 16,
        "obs_len_range": [0.1, 0.6],
        "n_obs": 0,
        "m": 0.0299,
        "Ixx": 1.395e-5,
        "Iyy": 1.395e-5,
        "Izz": 2.173e-5,
        "CT": 3.1582e-10,
        "CD": 7.9379e-12,
        "d": 0.03973,
    }

    fused_edge = False
    analytic_edge_jac = True  # single-backward fast path (edge_grad_to_state_jac)

    def __init__(self, num_agents, area_size, max_step=256, max_travel=None, dt=0.03,
                 params=None, device=None):
        # skip LinearDrone init (different dynamics); go to MultiAgentEnv
        from .base import MultiAgentEnv

        MultiAgentEnv.__init__(self, num_agents, area_size, max_step, max_travel, dt,
                               params, device)
        self._params.setdefault("car_radius", self._params["drone_radius"])
        self.normalize_by_CT = True
        self.vel_targets_scale = torch.tensor([2.0, 2.0, 0.5, 0.1])
        # hits per agent: min(16, beams grid size) (reference :76-77)
        self._n_hit = min(16, self._params["n_rays"] ** 2 // 2 + 2)
        self._K_ll_np = self._compute_K_ll()
        self._K_ll = torch.from_numpy(self._K_ll_np).float()
        self._K_nom_np = self._compute_K_nom()
        self._K_nom = torch.from_numpy(self._K_nom_np).float().to(self.device)

    # ---- dims -------------------------------------------------------------
    @property
    def state_dim(self) -> int:
        return 12

    @property
    def edge_dim(self) -> int:
        return 12

    @property
    def action_dim(self) -> int:
        return 4

    @property
    def n_rays(self) -> int:
        return self._n_hit

    @property
    def comm_radius(self) -> float:
        return self._params["comm_radius"]

    def state_lim(self, state=None) -> Tuple[Tensor, Tensor]:
        inf = math.inf
        pi4 = math.pi / 4
        lo = torch.tensor([-inf, -inf, -inf, -inf, -pi4, -pi4,
                           -0.3, -0.3, -0.3, -10.0, -10.0, -10.0])
        hi = torch.tensor([inf, inf, inf, inf, pi4, pi4,
                           0.3, 0.3, 0.3, 10.0, 10.0, 10.0])
        return lo, hi

    def action_lim(self) -> Tuple[Tensor, Tensor]:
        return -torch.ones(4), torch.ones(4)

    # ---- motor / dynamics helpers ------------------------------------------
    @property
    def u_eq(self) -> torch.Tensor:
        """Hover: total (CT-normalized) thrust = m g (reference :538-548)."""
        u = torch.full((4,), self._params["m"] * 9.81 / 4)
        if not self.normalize_by_CT:
            u = u / self._params["CT"]
        return u

    def _motor_mat(self) -> np.ndarray:
        """(w,p,q,r)-accel from motor thrusts (reference thrust_from_motor;
        note the reference uses Ixx for BOTH p and q terms in
        _single_agent_gu — kept faithful there, Iyy here as it writes)."""
        p = self._params
        m, Ixx, Iyy, Izz = p["m"], p["Ixx"], p["Iyy"], p["Izz"]
        CT, CD, d = p["CT"], p["CD"], p["d"]
        if self.normalize_by_CT:
            CT, CD = 1.0, CD / CT
        dw = CT * np.full(4, 1.0 / m)
        dp = CT * math.sqrt(2) * d * np.array([-1.0, -1.0, 1.0, 1.0]) / Ixx
        dq = CT * math.sqrt(2) * d * np.array([-1.0, 1.0, 1.0, -1.0]) / Iyy
        dr = CD * np.array([-1.0, 1.0, -1.0, 1.0]) / Izz
        return np.stack([dw, dp, dq, dr], axis=0)

    def _f_batched(self, st: Tensor) -> Tensor:
        """Drift dynamics f(x) (reference _single_agent_f, :305-351)."""
        p = self._params
        I = torch.tensor([p["Ixx"], p["Iyy"], p["Izz"]], device=st.device)
        phi, theta, psi = st[..., PHI], st[..., THETA], st[..., PSI]
        Rm = rotmat(phi, theta, psi)
        uvw = st[..., U : W + 1]  # slice: U,V,W = 6,7,8 (capture-safe)
        pqr = st[..., R_ : P_ + 1].flip(-1)  # [P_,Q_,R_] = [11,10,9]
        v_W = torch.einsum("...ij,...j->...i", Rm, uvw)
        c_phi, s_phi = torch.cos(phi), torch.sin(phi)
        c_th = torch.cos(theta)
        t_th = torch.tan(theta)
        zero = torch.zeros_like(phi)
        one = torch.ones_like(phi)
        # deuler (psi., theta., phi.) = mat @ pqr (reference rows :327-333)
        mat = torch.stack([
            torch.stack([zero, s_phi / c_th, c_phi / c_th], -1),
            torch.stack([zero, c_phi, -s_phi], -1),
            torch.stack([one, s_phi * t_th, c_phi * t_th], -1),
        ], dim=-2)
        deuler = torch.einsum("...ij,...j->...i", mat, pqr)
        acc_g = -Rm[..., 2, :] * 9.81  # body-frame gravity
        acc = -torch.cross(pqr, uvw, dim=-1) + acc_g
        pqr_dot = -torch.cross(pqr, I * pqr, dim=-1) / I
        rqp_dot = pqr_dot.flip(-1)
        return torch.cat([v_W, deuler, acc, rqp_dot], dim=-1)

    def _dev_const(self, name: str, build, device) -> Tensor:
        cache = getattr(self, "_const_cache", None)
        if cache is None:
            cache = self._const_cache = {}
        key = (name, str(device))
        if key not in cache:
            cache[key] = build().to(device)
        return cache[key]

    def _gu_batched(self, control: Tensor, device) -> Tensor:
        """Control contribution (reference _single_agent_gu): rows W,P,Q,R."""
        Mm = self._dev_const("Mm", lambda: torch.from_numpy(self._motor_mat()).float(), device)
        wpqr = control @ Mm.t()  # (..., 4): (w., p., q., r.)
        gu = torch.zeros(*control.shape[:-1], 12, device=device)
        gu[..., W] = wpqr[..., 0]
        gu[..., P_] = wpqr[..., 1]
        gu[..., Q_] = wpqr[..., 2]
        gu[..., R_] = wpqr[..., 3]
        return gu

    def _ll_state(self, st: Tensor) -> Tensor:
        """(phi, theta, psi, p, q, r, vx, vy, vz) world-frame (:550-562)."""
        Rm = rotmat(st[..., PHI], st[..., THETA], st[..., PSI])
        v_W = torch.einsum("...ij,...j->...i", Rm, st[..., U : W + 1])
        return torch.cat([st[..., PSI : PHI + 1].flip(-1), st[..., R_ : P_ + 1].flip(-1), v_W], dim=-1)

    def _ll_controls(self, st: Tensor, vel_targets: Tensor) -> Tensor:
        """Motor thrusts from the LL LQR (:564-576)."""
        ll = self._ll_state(st)
        des = torch.zeros_like(ll)
        des[..., 5] = vel_targets[..., 3]  # r target
        des[..., 6:9] = vel_targets[..., :3]
        K = self._dev_const("K_ll", lambda: self._K_ll.cpu(), st.device)
        ueq = self._dev_const("u_eq", lambda: self.u_eq, st.device)
        return ueq - torch.einsum("ij,...j->...i", K, ll - des)

    def _xdot_hl(self, st: Tensor, vel_targets_scaled: Tensor) -> Tensor:
        """High-level dynamics: scaled velocity targets -> LL LQR -> rigid
        body (:578-584)."""
        scale = self._dev_const("vts", lambda: self.vel_targets_scale.clone(), st.device)
        vt = self.clip_action(vel_targets_scaled) * scale
        control = self._ll_controls(st, vt)
        return self._f_batched(st) + self._gu_batched(control, st.device)

    def agent_xdot(self, agent_states: Tensor, action: Tensor) -> Tensor:
        return self._xdot_hl(agent_states, action)

    def agent_step_euler(self, agent_states: Tensor, action: Tensor) -> Tensor:
        # CrazyFlie integrates with RK4 (reference :620-625)
        return self.agent_step_rk4(agent_states, action)

    def agent_step_rk4(self, st: Tensor, action: Tensor) -> Tensor:
        dt = self._dt
        k1 = self._xdot_hl(st, action)
        k2 = self._xdot_hl(st + 0.5 * dt * k1, action)
        k3 = self._xdot_hl(st + 0.5 * dt * k2, action)
        k4 = self._xdot_hl(st + dt * k3, action)
        return self.clip_state(st + dt / 6.0 * (k1 + 2 * k2 + 2 * k3 + k4))

    def control_affine_dyn(self, state: Tensor) -> Tuple[Tensor, Tensor]:
        """f = xdot(x, 0); g = d xdot / d u at u = 0 (reference :423-430).

        The HL dynamics are affine in the (unclipped) targets:
        g = M_motor-rows . K_ll . d(ll_des)/d(vel) . diag(scale)."""
        f = self._xdot_hl(state, torch.zeros(*state.shape[:-1], 4, device=state.device))
        # d(motor)/d(vel_targets) = K_ll @ (-d des/d vt) with minus sign from
        # -(ll - des); then gu rows pick up M_motor columns
        K = self._K_ll_np  # (4, 9)
        Dsel = np.zeros((9, 4))
        Dsel[5, 3] = 1.0
        Dsel[6, 0] = Dsel[7, 1] = Dsel[8, 2] = 1.0
        du_motor = K @ Dsel  # (4 motors, 4 targets)
        Mm = self._motor_mat()  # (4 wpqr, 4 motors)
        g_wpqr = Mm @ du_motor * self.vel_targets_scale.numpy()[None, :]  # (4,4)
        g = torch.zeros(*state.shape[:-1], 12, 4, device=state.device)
        gt = torch.from_numpy(g_wpqr).float().to(state.device)
        g[..., W, :] = gt[0]
        g[..., P_, :] = gt[1]
        g[..., Q_, :] = gt[2]
        g[..., R_, :] = gt[3]
        return f, g

    # ---- LQR gains ----------------------------------------------------------
    def _xdot_ll_np(self, x: np.ndarray, u: np.ndarray) -> np.ndarray:
        """Low-level 9-state model for linearization (reference :353-421)."""
        p = self._params
        I = np.array([p["Ixx"], p["Iyy"], p["Izz"]])
        phi, theta, psi = x[0], x[1], x[2]
        pqr = x[3:6]
        mat = np.array([
            [1, math.sin(phi) * math.tan(theta), math.cos(phi) * math.tan(theta)],
            [0, math.cos(phi), -math.sin(phi)],
            [0, math.sin(phi) / math.cos(theta), math.cos(phi) / math.cos(theta)],
        ])
        deuler_rpy = mat @ pqr
        cph, sph = math.cos(phi), math.sin(phi)
        cth, sth = math.cos(theta), math.sin(theta)
        cps, sps = math.cos(psi), math.sin(psi)
        Rm = np.array([
            [cps * cth, cps * sth * sph - sps * cph, cps * sth * cph + sps * sph],
            [sps * cth, sps * sth * sph + cps * cph, sps * sth * cph - cps * sph],
            [-sth, cth * sph, cth * cph],
        ])
        pqr_dot = -np.cross(pqr, I * pqr) / I
        Mm = self._motor_mat()
        wpqr = Mm @ u
        acc_W = np.array([0.0, 0.0, -9.81]) + Rm @ np.array([0.0, 0.0, wpqr[0]])
        return np.concatenate([deuler_rpy, pqr_dot + wpqr[1:], acc_W])

    @staticmethod
    def _lqr_continuous(A, B, Q, R):
        import scipy.linalg

        S = scipy.linalg.solve_continuous_are(A, B, Q, R)
        return np.linalg.solve(R, B.T @ S)

    def _compute_K_ll(self) -> np.ndarray:
        u_eq = self.u_eq.numpy()

        def xdot(x, u):
            return self._xdot_ll_np(x, u + u_eq)

        x0, u0 = np.zeros(9), np.zeros(4)
        assert np.allclose(xdot(x0, u0), 0, atol=5e-5), "hover equilibrium check"
        eps = 1e-6
        A = np.stack([(xdot(x0 + eps * np.eye(9)[i], u0) - xdot(x0 - eps * np.eye(9)[i], u0))
                      / (2 * eps) for i in range(9)], axis=1)
        B = np.stack([(xdot(x0, u0 + eps * np.eye(4)[i]) - xdot(x0, u0 - eps * np.eye(4)[i]))
                      / (2 * eps) for i in range(4)], axis=1)
        # drop the psi row/col (index 2) as the reference does (:505-507)
        A = np.delete(np.delete(A, 2, axis=0), 2, axis=1)
        B = np.delete(B, 2, axis=0)
        Q = np.diag([1.0, 1.0, 1.0, 1.0, 1.0, 10.0, 10.0, 20.0])
        R_thrust = 0.01 * np.array([5.0, 1.0, 1.0, 1.0])
        T = self._motor_mat()
        R_motor = T.T @ np.diag(R_thrust) @ T
        K = self._lqr_continuous(A, B, Q, R_motor + 1e-9 * np.eye(4))
        return np.insert(K, 2, 0, axis=1)  # re-insert psi column of zeros

    def _xdot_hl_np(self, x: np.ndarray, u: np.ndarray) -> np.ndarray:
        """f64 single-agent HL dynamics for linearization."""
        p = self._params
        I = np.array([p["Ixx"], p["Iyy"], p["Izz"]])
        phi, theta, psi = x[PHI], x[THETA], x[PSI]
        cph, sph = math.cos(phi), math.sin(phi)
        cth, sth = math.cos(theta), math.sin(theta)
        cps, sps = math.cos(psi), math.sin(psi)
        tth = math.tan(theta)
        Rm = np.array([
            [cps * cth, cps * sth * sph - sps * cph, cps * sth * cph + sps * sph],
            [sps * cth, sps * sth * sph + cps * cph, sps * sth * cph - cps * sph],
            [-sth, cth * sph, cth * cph],
        ])
        uvw = x[[U, V, W]]
        pqr = x[[P_, Q_, R_]]
        v_W = Rm @ uvw
        mat = np.array([[0, sph / cth, cph / cth], [0, cph, -sph], [1, sph * tth, cph * tth]])
        deuler = mat @ pqr
        acc = -np.cross(pqr, uvw) - Rm[2, :] * 9.81
        pqr_dot = -np.cross(pqr, I * pqr) / I
        f = np.concatenate([v_W, deuler, acc, pqr_dot[::-1]])
        # LL controller
        vt = np.clip(u, -1, 1) * self.vel_targets_scale.numpy()
        ll = np.concatenate([[phi, theta, psi], pqr, v_W])
        des = np.zeros(9)
        des[5] = vt[3]
        des[6:9] = vt[:3]
        control = self.u_eq.numpy() - self._K_ll_np @ (ll - des)
        wpqr = self._motor_mat() @ control
        gu = np.zeros(12)
        gu[W], gu[P_], gu[Q_], gu[R_] = wpqr
        return f + gu

    def _compute_K_nom(self) -> np.ndarray:
        xdot = self._xdot_hl_np
        x0, u0 = np.zeros(12), np.zeros(4)
        eps = 1e-6
        A = np.stack([(xdot(x0 + eps * np.eye(12)[i], u0) - xdot(x0 - eps * np.eye(12)[i], u0))
                      / (2 * eps) for i in range(12)], axis=1)
        B = np.stack([(xdot(x0, u0 + eps * np.eye(4)[i]) - xdot(x0, u0 - eps * np.eye(4)[i]))
                      / (2 * eps) for i in range(4)], axis=1)
        Q = 2 * np.diag([50.0, 50.0, 50.0, 1, 1, 1, 1, 1, 1, 1, 1, 1.0])
        R = 4 * np.eye(4)
        return self._lqr_continuous(A, B, Q, R)

    # ---- graph / edges -------------------------------------------------------
    def _edge_states(self, states: Tensor) -> Tensor:
        """12-dim world-frame edge state: [pos, vel_W, z-axis_W, omega_W]
        (reference edge_state, :165-182)."""
        Rm = rotmat(states[..., PHI], states[..., THETA], states[..., PSI])
        v_W = torch.einsum("...ij,...j->...i", Rm, states[..., U : W + 1])
        z_W = Rm[..., :, 2]
        omega_W = torch.einsum("...ij,...j->...i", Rm, states[..., R_ : P_ + 1].flip(-1))
        return torch.cat([states[..., :3], v_W, z_W, omega_W], dim=-1)

    def edge_grad_to_state_jac(self, graph: GraphBatch, states: Tensor, ge: Tensor) -> Tensor:
        """Chain dh/d(edge_feats) -> dh_i/dx_j (M, N, N, 12) analytically:
        the DoubleIntegrator assembly (double_integrator.py
        edge_grad_to_state_jac) in EDGE-STATE space, then one extra chain
        through the per-agent 12x12 jacobian of the edge-state transform
        ``_edge_states`` (rotation-matrix features), obtained batched via
        torch.func.jacrev. Replaces N reverse passes through the whole GNN
        (reference gcbf_plus.py:310-317) with ONE backward + one tiny
        batched jacobian."""
        import torch.func as tf

        M, N, D, E = ge.shape
        n, r = self.num_agents, self.n_rays
        es = self._edge_states(states)
        recv = es[:, :n, None, :]
        senders = torch.cat(
            [
                es[:, None, :n].expand(M, n, n, E),
                es[:, n : 2 * n, None, :],
                es[:, 2 * n :].reshape(M, n, r, E),
            ],
            dim=2,
        )
        v = recv - senders
        p = v[..., :3]
        nrm = torch.sqrt(1e-6 + (p * p).sum(-1, keepdim=True))
        comm = self._params["comm_radius"]
        active = nrm > comm
        gp = ge[..., :3]
        gp_clip = comm * (gp / nrm - p * (gp * p).sum(-1, keepdim=True) / nrm.pow(3))
        gp_out = torch.where(active, gp_clip, gp)
        c = torch.cat([gp_out, ge[..., 3:]], dim=-1)  # cotangent wrt v (M,N,D,E)
        h_es = -c[:, :, :N, :].clone()  # wrt es of agent senders
        idx = torch.arange(N, device=states.device)
        h_es[:, idx, idx, :] += c.sum(dim=2)  # receiver side
        # chain through d(es)/d(state) for the agent nodes
        agent = states[:, :N].reshape(M * N, self.state_dim)
        JT = tf.vmap(tf.jacrev(self._edge_states))(agent)  # (M*N, 12, 12)
        JT = JT.reshape(M, N, E, self.state_dim)
        return torch.einsum("mije,mjes->mijs", h_es, JT)

    def edge_feats(self, graph: GraphBatch, states: Optional[Tensor] = None) -> Tensor:
        if states is None:
            states = graph.states
        es = self._edge_states(states)
        B, n, r = es.shape[0], self.num_agents, self.n_rays
        E = 12
        recv = es[:, :n, None, :]
        senders = torch.cat(
            [
                es[:, None, :n].expand(B, n, n, E),
                es[:, n : 2 * n, None, :],
                es[:, 2 * n :].reshape(B, n, r, E),
            ],
            dim=2,
        )
        e = recv - senders
        pos = e[..., :3]
        norm = torch.sqrt(1e-6 + (pos * pos).sum(-1, keepdim=True))
        comm = self._params["comm_radius"]
        coef = torch.where(norm > comm, comm / torch.clamp(norm, min=comm), torch.ones_like(norm))
        return torch.cat([pos * coef, e[..., 3:]], dim=-1)

    # ---- u_ref (reference u_ref_inner_single, :528-536) ----------------------
    def u_ref(self, graph: GraphBatch) -> Tensor:
        error = graph.agent_states - graph.goal_states
        dist = torch.linalg.vector_norm(error[..., :3], dim=-1, keepdim=True)
        coef = torch.where(dist > self.comm_radius,
                           self.comm_radius / dist.clamp_min(1e-4), torch.ones_like(dist))
        error = torch.cat([error[..., :3] * coef, error[..., 3:]], dim=-1)
        u = -torch.einsum("ij,...j->...i", self._K_nom.to(error.device), error)
        return self.clip_action(u)

    def step(self, graph: GraphBatch, action: Tensor) -> StepResult:
        action = self.clip_action(action)
        next_agent = self.agent_step_rk4(graph.agent_states, action)
        reward = -((action - self.u_ref(graph)).square().sum(-1)).mean(-1)
        cost = self.get_cost(graph)
        done = torch.zeros(graph.batch_size, dtype=torch.bool, device=graph.device)
        next_graph = self.get_graph(next_agent, graph.goal_states, graph.env_states)
        return StepResult(next_graph, reward, cost, done, {})

    def forward_graph(self, graph: GraphBatch, action: Tensor) -> GraphBatch:
        action = self.clip_action(action)
        return graph.with_agent_states(self.agent_step_rk4(graph.agent_states, action))

    # lidar: 3D fan over params['n_rays'] beams, top-n_hit returns
    def get_lidar_hits(self, agent_pos: Tensor, obstacles: Sphere) -> Tensor:
        from .utils import get_lidar
        from .. import ops

        if agent_pos.is_cuda and ops.hip_available() and obstacles.n_obs > 0:
            return ops.raytrace_sphere_topk(
                agent_pos, obstacles.center, obstacles.radius,
                self._params["n_rays"], self._n_hit,
                self._params["comm_radius"])
        return get_lidar(agent_pos, obstacles, self._params["n_rays"],
                         self._params["comm_radius"], max_returns=self._n_hit)
