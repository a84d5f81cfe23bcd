"""PPO (clip surrogate + value clip + entropy bonus, GAE).

Reference parity: ``agilerl/algorithms/ppo.py:56`` — clip losses
(:800-832), GAE from the rollout buffer, recurrent BPTT path, target-KL
early stop.  The GAE scan and (on GPU) the policy-loss reductions run as
HIP kernels via ``agilerl_amd.ops``.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch
import torch.nn as nn

from ..components.rollout_buffer import RolloutBuffer
from ..networks.actors import StochasticActor
from ..networks.value_networks import ValueNetwork
from ..spaces import Space
from .core.base import RLAlgorithm
from .core.optimizer_wrapper import OptimizerWrapper
from .core.registry import HyperparameterConfig, NetworkGroup, OptimizerConfig, RLParameter

__all__ = ["PPO"]


def default_hp_config() -> HyperparameterConfig:
    return HyperparameterConfig(
        lr=RLParameter(min=1e-5, max=1e-2),
        batch_size=RLParameter(min=64, max=4096, dtype=int),
        clip_coef=RLParameter(min=0.05, max=0.4),
        ent_coef=RLParameter(min=1e-4, max=0.05),
        update_epochs=RLParameter(min=1, max=10, dtype=int),
    )


class PPO(RLAlgorithm):
    def __init__(
        self,
        observation_space: Space,
        action_space: Space,
        index: int = 0,
        hp_config: Optional[HyperparameterConfig] = None,
        net_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        batch_size: int = 512,
        lr: float = 3e-4,
        learn_step: int = 128,
        gamma: float = 0.99,
        gae_lambda: float = 0.95,
        clip_coef: float = 0.2,
        clip_vloss: bool = True,
        ent_coef: float = 0.01,
        vf_coef: float = 0.5,
        update_epochs: int = 4,
        max_grad_norm: float = 0.5,
        target_kl: Optional[float] = None,
        normalize_advantage: bool = True,
        log_std_init: float = 0.0,
        action_std_init: Optional[float] = None,
        latent_dim: int = 64,
        recurrent: bool = False,
        num_envs: int = 1,
        max_seq_len: Optional[int] = None,
        bptt_sequence_type: str = "chunked",
        rollout_buffer_config: Optional[Dict[str, Any]] = None,
        share_encoders: bool = False,
        actor_network=None,
        critic_network=None,
        device: str = "cpu",
        **kwargs,
    ):
        super().__init__(
            observation_space,
            action_space,
            index=index,
            learn_step=learn_step,
            device=device,
            hp_config=hp_config or default_hp_config(),
            name="PPO",
        )
        self._accept_compat_kwargs(**kwargs)
        self.batch_size = int(batch_size)
        self.lr = float(lr)
        self.gamma = float(gamma)
        self.gae_lambda = float(gae_lambda)
        self.clip_coef = float(clip_coef)
        self.clip_vloss = bool(clip_vloss)
        self.ent_coef = float(ent_coef)
        self.vf_coef = float(vf_coef)
        self.update_epochs = int(update_epochs)
        self.max_grad_norm = float(max_grad_norm)
        self.target_kl = target_kl
        self.normalize_advantage = normalize_advantage
        if action_std_init is not None:
            # reference ppo.py:143 `action_std_init` — despite the name it is
            # the initial LOG standard deviation (docstring ppo.py:279)
            log_std_init = float(action_std_init)
        self.action_std_init = action_std_init
        self.log_std_init = float(log_std_init)
        self.net_config = net_config
        self.latent_dim = latent_dim
        self.recurrent = recurrent
        # reference ppo.py:154-161: num_envs sizes the rollout buffer;
        # max_seq_len = truncated-BPTT window; bptt_sequence_type picks the
        # window stride (typing.py:473); rollout_buffer_config is forwarded
        # to RolloutBuffer by the training loop
        self.num_envs = int(num_envs)
        self.sequence_length = int(max_seq_len) if max_seq_len else 16
        if bptt_sequence_type not in ("chunked", "maximum", "fifty_percent_overlap"):
            raise ValueError(
                "bptt_sequence_type must be chunked | maximum | fifty_percent_overlap"
            )
        self.bptt_sequence_type = bptt_sequence_type
        self.rollout_buffer_config = dict(rollout_buffer_config or {})
        if recurrent and (net_config is None or net_config.get("arch") != "lstm"):
            net_config = dict(net_config or {})
            net_config["arch"] = "lstm"
            net_config.setdefault("hidden_state_size", 64)
            self.net_config = net_config

        if actor_network is not None:
            # user-supplied policy net (reference ppo.py actor_network):
            # maps preprocessed obs -> distribution head outputs
            from ..networks.base import CustomStochasticAdapter

            self.actor = CustomStochasticAdapter(
                actor_network, observation_space, action_space, device=device
            )
        else:
            self.actor = StochasticActor(
                observation_space,
                action_space,
                encoder_config=net_config,
                head_config=head_config,
                latent_dim=latent_dim,
                log_std_init=log_std_init,
                device=device,
            )
        if critic_network is not None:
            # user-supplied value net: preprocessed obs -> (B,1) value
            # (reference ppo.py critic_network)
            from ..networks.base import CustomNetworkAdapter

            self.critic = CustomNetworkAdapter(critic_network, observation_space,
                                               device=device)
        else:
            self.critic = ValueNetwork(
                observation_space,
                encoder_config=net_config,
                head_config=head_config,
                latent_dim=latent_dim,
                device=device,
            )
        if (actor_network is not None or critic_network is not None) and share_encoders:
            raise ValueError("share_encoders is not supported with custom networks")
        if actor_network is not None and recurrent:
            raise ValueError("recurrent=True is not supported with a custom actor_network")
        self.share_encoders = bool(share_encoders)
        if self.share_encoders:
            self.critic.encoder = self.actor.encoder  # one encoder, two heads
        self.optimizer = OptimizerWrapper(
            torch.optim.Adam, [self.actor, self.critic], lr=self.lr
        )

        self.register_network_group(NetworkGroup(eval_network="actor", policy=True))
        self.register_network_group(NetworkGroup(eval_network="critic"))
        self.register_optimizer(
            OptimizerConfig(name="optimizer", networks=["actor", "critic"], lr_name="lr")
        )
        self.register_mutation_hook("_clear_learn_graph")
        self._learn_graph = None
        self._learn_static = None

    def _clear_learn_graph(self) -> None:
        self._learn_graph = None
        self._learn_static = None

    def post_clone_hook(self, parent) -> None:
        if getattr(self, "share_encoders", False):
            self.critic.encoder = self.actor.encoder
            self._reinit_optimizers()

    def _apply_checkpoint(self, ckpt) -> None:
        super()._apply_checkpoint(ckpt)
        if getattr(self, "share_encoders", False):
            self.critic.encoder = self.actor.encoder
            self._reinit_optimizers()

    # ------------------------------------------------------------------
    def get_action(
        self,
        obs,
        action_mask: Optional[np.ndarray] = None,
        training: bool = True,
    ):
        """Returns (action_np, log_prob, entropy, value) during training,
        or action_np when ``training=False`` (greedy)."""
        mask_t = (
            torch.as_tensor(np.asarray(action_mask), device=self.device)
            if action_mask is not None
            else None
        )
        with torch.no_grad():
            pre = self.actor.preprocess(obs)
            if not training:
                return self.actor.deterministic_action(pre, mask_t).cpu().numpy()
            action, log_prob, entropy = self.actor.sample(pre, mask_t)
            value = self.critic(self.critic.preprocess(obs)).squeeze(-1)
        return action.cpu().numpy(), log_prob, entropy, value

    def get_values(self, obs) -> torch.Tensor:
        with torch.no_grad():
            return self.critic(self.critic.preprocess(obs)).squeeze(-1)

    # ------------------------------------------------------------------
    # Recurrent (BPTT) path
    # ------------------------------------------------------------------
    def init_hidden(self, batch_size: int) -> Dict[str, torch.Tensor]:
        ha, ca = self.actor.initial_hidden(batch_size)
        hc, cc = self.critic.initial_hidden(batch_size)
        return {"ha": ha, "ca": ca, "hc": hc, "cc": cc}

    def get_action_recurrent(
        self, obs, hidden: Dict[str, torch.Tensor], training: bool = True
    ):
        with torch.no_grad():
            head_out, (ha, ca) = self.actor.forward_step(obs, (hidden["ha"], hidden["ca"]))
            if not training:
                action = self.actor.dist_layer.mode(head_out)
                return action.cpu().numpy(), {"ha": ha, "ca": ca, "hc": hidden["hc"], "cc": hidden["cc"]}
            action, log_prob, _ent = self.actor.dist_layer.sample(head_out)
            v_out, (hc, cc) = self.critic.forward_step(obs, (hidden["hc"], hidden["cc"]))
        new_hidden = {"ha": ha, "ca": ca, "hc": hc, "cc": cc}
        return action.cpu().numpy(), log_prob, v_out.squeeze(-1), new_hidden

    def _learn_recurrent(self, rollout: RolloutBuffer) -> Dict[str, float]:
        stats = {"policy_loss": 0.0, "value_loss": 0.0, "entropy": 0.0, "approx_kl": 0.0}
        n = 0
        L = self.sequence_length
        for _ in range(self.update_epochs):
            seq_bs = max(self.batch_size // L, 1)
            for mb in rollout.get_sequence_minibatches(
                L, seq_bs, sequence_type=self.bptt_sequence_type
            ):
                obs_seq = mb["obs"].float()  # (B, L, F)
                B = obs_seq.shape[0]
                hs = mb["hidden_state"]
                h0a = (hs["ha"][:, 0].transpose(0, 1).contiguous(), hs["ca"][:, 0].transpose(0, 1).contiguous())
                h0c = (hs["hc"][:, 0].transpose(0, 1).contiguous(), hs["cc"][:, 0].transpose(0, 1).contiguous())
                head_seq = self.actor.forward_sequence(obs_seq, h0a)
                v_seq = self.critic.forward_sequence(obs_seq, h0c).reshape(B * L)
                flat_out = head_seq.reshape(B * L, -1)
                actions = mb["action"].reshape(B * L, *mb["action"].shape[2:])
                log_prob, entropy = self.actor.dist_layer.log_prob_entropy(flat_out, actions)

                adv = mb["advantages"].reshape(-1)
                if self.normalize_advantage and adv.numel() > 1:
                    adv = (adv - adv.mean()) / (adv.std() + 1e-8)
                old_lp = mb["log_prob"].reshape(-1)
                returns = mb["returns"].reshape(-1)
                old_values = mb["value"].reshape(-1)
                log_ratio = log_prob.reshape(-1) - old_lp
                ratio = log_ratio.exp()
                pg = torch.maximum(
                    -adv * ratio, -adv * ratio.clamp(1 - self.clip_coef, 1 + self.clip_coef)
                ).mean()
                if self.clip_vloss:
                    v_clip = old_values + (v_seq - old_values).clamp(-self.clip_coef, self.clip_coef)
                    vloss = 0.5 * torch.maximum((v_seq - returns) ** 2, (v_clip - returns) ** 2).mean()
                else:
                    vloss = 0.5 * ((v_seq - returns) ** 2).mean()
                loss = pg + self.vf_coef * vloss - self.ent_coef * entropy.mean()
                self.optimizer.zero_grad()
                loss.backward()
                nn.utils.clip_grad_norm_(
                    [p for net in (self.actor, self.critic) for p in net.parameters()],
                    self.max_grad_norm,
                )
                self.optimizer.step()
                with torch.no_grad():
                    stats["approx_kl"] += float(((ratio - 1) - log_ratio).mean())
                stats["policy_loss"] += float(pg.detach())
                stats["value_loss"] += float(vloss.detach())
                stats["entropy"] += float(entropy.mean().detach())
                n += 1
        return {k: v / max(n, 1) for k, v in stats.items()}

    def get_action_device(self, obs_t: torch.Tensor, training: bool = True):
        """Device-native action path (TorchVecEnv): tensors in, tensors out —
        no host round-trips in the collect loop."""
        with torch.no_grad():
            pre = self.actor.preprocess(obs_t)
            if not training:
                return self.actor.deterministic_action(pre)
            action, log_prob, entropy = self.actor.sample(pre)
            value = self.critic(self.critic.preprocess(obs_t)).squeeze(-1)
        return action, log_prob, entropy, value

    # ------------------------------------------------------------------
    def learn(self, rollout) -> Dict[str, float]:
        """``rollout``: a RolloutBuffer, or a flat dict of (T*N, ...) tensors
        (the hipGraph collector's output) with advantages/returns included."""
        if self.recurrent and not isinstance(rollout, dict):
            return self._learn_recurrent(rollout)
        if isinstance(rollout, dict):
            from ..spaces import Discrete as _Discrete

            if (
                torch.cuda.is_available()
                and rollout["obs"].is_cuda
                and isinstance(self.action_space, _Discrete)
                and self.target_kl is None
            ):
                return self._graphed_flat_learn(rollout)
            minibatches = lambda: self._flat_minibatches(rollout)
        else:
            assert rollout.advantages is not None, "call compute_returns_and_advantages first"
            minibatches = lambda: rollout.get_minibatches(self.batch_size)
        stats = {"policy_loss": 0.0, "value_loss": 0.0, "entropy": 0.0, "approx_kl": 0.0}
        n_updates = 0
        early_stop = False
        for _ in range(self.update_epochs):
            if early_stop:
                break
            for mb in minibatches():
                loss_stats = self._update_minibatch(mb)
                for k in stats:
                    stats[k] += loss_stats[k]
                n_updates += 1
                if self.target_kl is not None and loss_stats["approx_kl"] > 1.5 * self.target_kl:
                    early_stop = True
                    break
        if n_updates:
            stats = {k: v / n_updates for k, v in stats.items()}
        return stats

    def test(self, env, max_steps: Optional[int] = None, loop: int = 3, **kwargs) -> float:
        if not self.recurrent:
            return super().test(env, max_steps=max_steps, loop=loop, **kwargs)
        with torch.no_grad():
            rewards = []
            for _ in range(loop):
                obs, _ = env.reset()
                hidden = self.init_hidden(env.num_envs)
                done_mask = np.zeros(env.num_envs, dtype=bool)
                ep_rew = np.zeros(env.num_envs)
                steps = 0
                while not done_mask.all():
                    action, hidden = self.get_action_recurrent(obs, hidden, training=False)
                    obs, rew, term, trunc, _ = env.step(action)
                    ep_rew += np.asarray(rew) * (~done_mask)
                    done_mask |= np.asarray(term) | np.asarray(trunc)
                    steps += 1
                    if max_steps is not None and steps >= max_steps:
                        break
                rewards.append(ep_rew.mean())
        fitness = float(np.mean(rewards))
        self.fitness.append(fitness)
        return fitness

    def _flat_minibatches(self, flat: Dict[str, torch.Tensor]):
        n = flat["advantages"].shape[0]
        idx = torch.randperm(n, device=flat["advantages"].device)
        for start in range(0, n, self.batch_size):
            sel = idx[start : start + self.batch_size]
            yield {k: v[sel] for k, v in flat.items()}

    # ------------------------------------------------------------------
    # hipGraph-captured update over flat rollouts (discrete actions).
    # One graph replay per minibatch; stats read once per learn() call.
    # ------------------------------------------------------------------
    _GRAPH_KEYS = ("obs", "action", "log_prob", "advantages", "returns", "value")

    def _graph_update_body(self, st: Dict[str, torch.Tensor]) -> None:
        adv = st["advantages"]
        if self.normalize_advantage:
            adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        logits = self.actor(self.actor.preprocess(st["obs"]))
        logp_all = torch.log_softmax(logits, dim=-1)
        new_log_prob = logp_all.gather(1, st["action"].unsqueeze(1)).squeeze(1)
        entropy = -(logp_all.exp() * logp_all).sum(-1)
        values = self.critic(self.critic.preprocess(st["obs"])).reshape(-1)

        log_ratio = new_log_prob - st["log_prob"]
        ratio = log_ratio.exp()
        pg1 = -adv * ratio
        pg2 = -adv * ratio.clamp(1 - self.clip_coef, 1 + self.clip_coef)
        policy_loss = torch.maximum(pg1, pg2).mean()
        if self.clip_vloss:
            v_clipped = st["value"] + (values - st["value"]).clamp(-self.clip_coef, self.clip_coef)
            value_loss = 0.5 * torch.maximum(
                (values - st["returns"]) ** 2, (v_clipped - st["returns"]) ** 2
            ).mean()
        else:
            value_loss = 0.5 * ((values - st["returns"]) ** 2).mean()
        entropy_loss = entropy.mean()
        loss = policy_loss + self.vf_coef * value_loss - self.ent_coef * entropy_loss
        self.optimizer.zero_grad(set_to_none=False)
        loss.backward()
        nn.utils.clip_grad_norm_(
            [p for net in (self.actor, self.critic) for p in net.parameters()],
            self.max_grad_norm, foreach=True,
        )
        self.optimizer.step()
        st["stats"][0] += policy_loss.detach()
        st["stats"][1] += value_loss.detach()
        st["stats"][2] += entropy_loss.detach()
        st["stats"][3] += ((ratio - 1) - log_ratio).mean().detach()

    def _graphed_flat_learn(self, flat: Dict[str, torch.Tensor]) -> Dict[str, float]:
        n = flat["advantages"].shape[0]
        B = min(self.batch_size, n)
        if self._learn_static is not None and self._learn_static["obs"].shape[0] != B:
            self._clear_learn_graph()
        if self._learn_graph is None:
            prev_opt = self.optimizer.optimizer
            prev_state = dict(prev_opt.state) if prev_opt is not None else {}
            self.optimizer.optimizer = torch.optim.Adam(
                [p for net in (self.actor, self.critic) for p in net.parameters()
                 if p.requires_grad],
                lr=self.lr, capturable=True,
            )
            st = {
                "obs": flat["obs"][:B].clone().float(),
                "action": flat["action"][:B].clone().long(),
                "log_prob": flat["log_prob"][:B].clone().float(),
                "advantages": flat["advantages"][:B].clone().float(),
                "returns": flat["returns"][:B].clone().float(),
                "value": flat["value"][:B].clone().float(),
                "stats": torch.zeros(4, device=flat["obs"].device),
            }
            self._learn_static = st
            # warmup + capture run REAL updates; snapshot and restore so the
            # first minibatch replays start from the pre-capture weights
            nets = [p for net in (self.actor, self.critic) for p in net.parameters()]
            saved = [p.detach().clone() for p in nets]
            torch.cuda.synchronize()
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(3):
                    self._graph_update_body(st)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            self._learn_graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self._learn_graph):
                self._graph_update_body(st)
            with torch.no_grad():
                for p, sv in zip(nets, saved):
                    p.copy_(sv)
                # restore optimizer moments: carry over any pre-existing Adam
                # state (checkpoint-loaded or accumulated eagerly) into the
                # capturable optimizer's state tensors; params without prior
                # state start from zeroed moments
                for p, group_state in self.optimizer.optimizer.state.items():
                    prior = prev_state.get(p)
                    for k, v in group_state.items():
                        if not torch.is_tensor(v):
                            continue
                        pv = prior.get(k) if prior else None
                        if pv is not None and torch.is_tensor(pv) and pv.shape == v.shape:
                            v.copy_(pv.to(v.device, v.dtype))
                        elif pv is not None and k == "step" and v.numel() == 1:
                            v.fill_(float(pv))
                        else:
                            v.zero_()
            torch.cuda.synchronize()
        st = self._learn_static
        st["stats"].zero_()
        n_updates = 0
        n_full = (n // B) * B
        for _ in range(self.update_epochs):
            perm = torch.randperm(n, device=st["obs"].device)
            for start in range(0, n_full, B):
                sel = perm[start : start + B]
                for key in self._GRAPH_KEYS:
                    st[key].copy_(
                        flat[key].index_select(0, sel).to(st[key].dtype), non_blocking=True
                    )
                self._learn_graph.replay()
                n_updates += 1
        stats_v = (st["stats"] / max(n_updates, 1)).cpu()
        return {
            "policy_loss": float(stats_v[0]),
            "value_loss": float(stats_v[1]),
            "entropy": float(stats_v[2]),
            "approx_kl": float(stats_v[3]),
        }

    def _update_minibatch(self, mb: Dict[str, torch.Tensor]) -> Dict[str, float]:
        obs = mb["obs"]
        actions = mb["action"].to(self.device)
        old_log_prob = mb["log_prob"].to(self.device).reshape(-1)
        advantages = mb["advantages"].to(self.device).reshape(-1)
        returns = mb["returns"].to(self.device).reshape(-1)
        old_values = mb["value"].to(self.device).reshape(-1)
        mask = mb.get("action_mask")

        if self.normalize_advantage and advantages.numel() > 1:
            advantages = (advantages - advantages.mean()) / (advantages.std() + 1e-8)

        new_log_prob, entropy = self.actor.evaluate_actions(
            self.actor.preprocess(obs), actions, mask
        )
        values = self.critic(self.critic.preprocess(obs)).reshape(-1)

        log_ratio = new_log_prob.reshape(-1) - old_log_prob
        ratio = log_ratio.exp()
        with torch.no_grad():
            approx_kl = ((ratio - 1) - log_ratio).mean()

        pg1 = -advantages * ratio
        pg2 = -advantages * ratio.clamp(1 - self.clip_coef, 1 + self.clip_coef)
        policy_loss = torch.maximum(pg1, pg2).mean()

        if self.clip_vloss:
            v_clipped = old_values + (values - old_values).clamp(
                -self.clip_coef, self.clip_coef
            )
            value_loss = 0.5 * torch.maximum(
                (values - returns) ** 2, (v_clipped - returns) ** 2
            ).mean()
        else:
            value_loss = 0.5 * ((values - returns) ** 2).mean()

        entropy_loss = entropy.mean()
        loss = policy_loss + self.vf_coef * value_loss - self.ent_coef * entropy_loss

        self.optimizer.zero_grad()
        loss.backward()
        nn.utils.clip_grad_norm_(
            [p for net in (self.actor, self.critic) for p in net.parameters()],
            self.max_grad_norm,
        )
        self.optimizer.step()
        return {
            "policy_loss": float(policy_loss.detach()),
            "value_loss": float(value_loss.detach()),
            "entropy": float(entropy_loss.detach()),
            "approx_kl": float(approx_kl),
        }
