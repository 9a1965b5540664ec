"""Pure-PyTorch (fp32, eager) reference implementations of every fused HIP op.

These are the numerics oracles the HIP kernels (distributed_rl_amd/ops/hip/*)
are unit-tested against, and the CPU execution path (no GPU present).
Each function documents the reference code whose math it reproduces
(file:line into /root/reference/).
"""

from __future__ import annotations

from typing import Tuple

import torch

# ---------------------------------------------------------------------------
# K1 — uint8 -> float /255 dequant (APE_X/Learner.py:61-67 does
#      torch.tensor(u8).float()/255 on the GPU synchronously)
# ---------------------------------------------------------------------------


def dequant_frames(x_u8: torch.Tensor, dtype: torch.dtype = torch.float32) -> torch.Tensor:
    return x_u8.to(dtype) / 255.0


# ---------------------------------------------------------------------------
# K4 — n-step double-DQN TD loss + new priority + IS-weighted loss
#      (APE_X/Learner.py:83-114; gamma**n with done mask; TD clip to [-1,1];
#      priority (|clip(td)|+1e-7)**alpha; loss 0.5*mean(w*td^2))
# ---------------------------------------------------------------------------


def nstep_dqn_loss(
    q_online_s: torch.Tensor,  # (B, A) differentiable
    q_online_sp: torch.Tensor,  # (B, A) no-grad
    q_target_sp: torch.Tensor,  # (B, A) no-grad
    actions: torch.Tensor,  # (B,) int64
    rewards: torch.Tensor,  # (B,) discounted n-step return
    dones: torch.Tensor,  # (B,) float 1.0 where terminal
    weights: torch.Tensor,  # (B,) PER importance weights
    gamma: float,
    n_step: int,
    alpha: float,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (loss, new_priorities).

    Note the reference hardcodes gamma=0.99 in the bootstrap term
    (APE_X/Learner.py:103) regardless of cfg GAMMA — a defect per SURVEY §7;
    we use the configured gamma.
    """
    B = q_online_s.shape[0]
    a_star = q_online_sp.argmax(dim=1)  # double-DQN action selection
    next_q = q_target_sp.gather(1, a_star.unsqueeze(1)).squeeze(1)
    target = rewards + (gamma ** n_step) * next_q * (1.0 - dones)
    q_sa = q_online_s.gather(1, actions.long().unsqueeze(1)).squeeze(1)
    td = (target.detach() - q_sa).clamp(-1.0, 1.0)
    new_priority = (td.detach().abs() + 1e-7) ** alpha
    loss = 0.5 * (weights * td.pow(2)).mean()
    return loss, new_priority


# ---------------------------------------------------------------------------
# K6 — R2D2 value rescaling h / h^-1 (R2D2/Learner.py:22-35)
# ---------------------------------------------------------------------------

_RESCALE_EPS = 1e-3


def value_rescale(x: torch.Tensor, eps: float = _RESCALE_EPS) -> torch.Tensor:
    return torch.sign(x) * ((x.abs() + 1.0).sqrt() - 1.0) + eps * x


def inv_value_rescale(x: torch.Tensor, eps: float = _RESCALE_EPS) -> torch.Tensor:
    # closed-form inverse (R2D2/Learner.py:29-35)
    return torch.sign(x) * (
        ((1.0 + 4.0 * eps * (x.abs() + 1.0 + eps)).sqrt() - 1.0).pow(2)
        / (4.0 * eps ** 2)
        - 1.0
    )


# ---------------------------------------------------------------------------
# K7 — R2D2 sequence priority: eta-mix 0.9*max + 0.1*mean over per-step |td|
#      then **alpha (R2D2/Learner.py:175-181, R2D2/Player.py:209-211)
# ---------------------------------------------------------------------------


def sequence_priority(
    td_abs: torch.Tensor, alpha: float, eta: float = 0.9
) -> torch.Tensor:
    """td_abs: (T, B) -> (B,) priorities."""
    mix = eta * td_abs.max(dim=0).values + (1.0 - eta) * td_abs.mean(dim=0)
    return mix ** alpha


# ---------------------------------------------------------------------------
# K8 — IMPALA V-trace (IMPALA/Learner.py:121-226)
# ---------------------------------------------------------------------------


def vtrace(
    behavior_log_prob: torch.Tensor,  # (T, B) log mu(a|s)
    target_log_prob: torch.Tensor,  # (T, B) log pi(a|s)
    rewards: torch.Tensor,  # (T, B)
    values: torch.Tensor,  # (T, B) V(s_t) under target net
    bootstrap_value: torch.Tensor,  # (B,) V(s_T)
    not_done: torch.Tensor,  # (B,) 1.0 if trajectory did NOT terminate
    gamma: float,
    rho_bar: float = 1.0,
    c_bar: float = 1.0,
    lam: float = 1.0,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (vs, pg_advantage, rho_clipped).

    Terminal handling matches the reference: the bootstrap value is masked
    by not_done (IMPALA/Player.py:183-186 emits not_done; Learner applies it
    to the bootstrap), and the scan runs reversed in time with
    c_i = lam * min(c_bar, rho_i) (IMPALA/Learner.py:176-200).

    DELIBERATE DIVERGENCE (also recorded in docs/PARITY.md): the reference
    omits the clipped-rho factor on the delta of the LAST unrolled step
    (IMPALA/Learner.py:176-185 applies rho only inside the loop body, not on
    the step that seeds the reversed scan); we apply rho_clipped to EVERY
    step's delta — the V-trace paper's form. Values differ from the
    reference whenever pi != mu on the final step of an unroll.
    """
    T, B = rewards.shape
    rho = (target_log_prob - behavior_log_prob).exp()
    rho_c = rho.clamp(max=rho_bar)
    c = lam * rho.clamp(max=c_bar)

    values_tp1 = torch.cat(
        [values[1:], (bootstrap_value * not_done).unsqueeze(0)], dim=0
    )
    deltas = rho_c * (rewards + gamma * values_tp1 - values)

    acc = torch.zeros(B, dtype=values.dtype, device=values.device)
    vs_minus_v = torch.empty_like(values)
    for t in reversed(range(T)):
        acc = deltas[t] + gamma * c[t] * acc
        vs_minus_v[t] = acc
    vs = values + vs_minus_v

    vs_tp1 = torch.cat([vs[1:], (bootstrap_value * not_done).unsqueeze(0)], dim=0)
    pg_adv = rho_c * (rewards + gamma * vs_tp1 - values)
    return vs, pg_adv, rho_c


# ---------------------------------------------------------------------------
# K9 — IMPALA policy/critic losses (IMPALA/Learner.py:95-119, 224)
# ---------------------------------------------------------------------------


def impala_loss(
    logits: torch.Tensor,  # (T*B, A) differentiable (learner policy head)
    values_pred: torch.Tensor,  # (T*B,) differentiable critic
    actions: torch.Tensor,  # (T*B,) int64
    pg_adv: torch.Tensor,  # (T*B,) no-grad
    vs: torch.Tensor,  # (T*B,) no-grad V-trace targets
    entropy_coef: float,
    critic_coef: float = 1.0,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    log_pi = torch.log_softmax(logits, dim=-1)
    pi = log_pi.exp()
    entropy = -(pi * log_pi).sum(-1).mean()
    log_pi_a = log_pi.gather(1, actions.unsqueeze(1)).squeeze(1)
    obj_actor = (log_pi_a * pg_adv.detach()).mean() + entropy_coef * entropy
    critic_loss = 0.5 * torch.nn.functional.mse_loss(values_pred, vs.detach())
    total = -obj_actor + critic_coef * critic_loss
    return total, obj_actor, critic_loss, entropy


# ---------------------------------------------------------------------------
# K10 host oracle — proportional PER semantics (contract from SURVEY §2.8):
#   P(i) = p_i / sum(p)   (alpha applied producer-side),
#   IS weight w_i = (1/(n*P_i))^beta / max_j w_j   (APE_X/ReplayMemory.py:64-67)
# ---------------------------------------------------------------------------


def per_probabilities(priorities: torch.Tensor) -> torch.Tensor:
    return priorities / priorities.sum()


def per_is_weights(
    probs: torch.Tensor, n: int, beta: float
) -> torch.Tensor:
    w = (1.0 / (n * probs)) ** beta
    return w / w.max()


# ---------------------------------------------------------------------------
# n-step return folding (actor side; APE_X/Player.py:38-51)
# ---------------------------------------------------------------------------


def fold_nstep_reward(rewards: torch.Tensor, gamma: float) -> torch.Tensor:
    """rewards: (..., n) -> (...) discounted sum_t gamma^t r_t."""
    n = rewards.shape[-1]
    disc = gamma ** torch.arange(n, dtype=rewards.dtype, device=rewards.device)
    return (rewards * disc).sum(-1)
