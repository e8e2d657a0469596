"""PPO method config, losses, KL controllers, and value-head/hydra wrappers.

Parity target: reference trlx/models/modeling_ppo.py —
AdaptiveKLController/FixedKLController (35-67), PPOConfig with GAE (136-173)
and the clipped PPO loss (175-238), AutoModelForCausalLMWithValueHead
(266-382), AutoModelForCausalLMWithHydraValueHead + ModelBranch (385-544).

MI355X redesign notes:
- GAE runs as the wave-parallel affine-scan HIP kernel (ops.gae) instead of a
  Python reverse loop; whitening uses the fused stats kernels + one RCCL
  all-reduce when distributed.
- The hydra frozen branch re-runs only the top unfrozen blocks from the
  trunk's stashed hidden state: ``forward(..., return_ref_logits=True)`` gives
  policy logits, values, AND reference logits in ONE trunk pass (the reference
  needs forward + forward_hydra = two).  The frozen bottom layers are shared
  storage — 288 GB HBM keeps the whole hydra resident.
- No per-arch branch classes: the native Block is arch-agnostic.
"""

import copy
from dataclasses import dataclass, field
from typing import Any, Dict, Optional, Tuple

import torch
import torch.nn as nn

from .. import ops
from ..data.method_configs import MethodConfig, register_method
from ..utils.modeling import flatten_dict, get_tensor_stats, make_head, whiten
from .modeling_base import PreTrainedModelWrapper
from .nn.generation import generate
from .nn.transformer import CausalTransformer


class AdaptiveKLController:
    """Adaptive KL controller per Ziegler et al. "Fine-Tuning Language Models
    from Human Preferences" (reference modeling_ppo.py:35-53)."""

    def __init__(self, init_kl_coef: float, target: float, horizon: int):
        self.value = init_kl_coef
        self.target = target
        self.horizon = horizon

    def update(self, current: float, n_steps: int):
        proportional_error = max(min(current / self.target - 1, 0.2), -0.2)
        mult = 1 + proportional_error * n_steps / self.horizon
        self.value *= mult


class FixedKLController:
    """Fixed KL coefficient (reference modeling_ppo.py:56-67)."""

    def __init__(self, kl_coef: float):
        self.value = kl_coef

    def update(self, current: float, n_steps: int):
        pass


@dataclass
@register_method
class PPOConfig(MethodConfig):
    """PPO hyperparameters (reference modeling_ppo.py:73-238).

    :param ppo_epochs: optimization epochs per rollout batch
    :param num_rollouts: experiences to collect per outer epoch
    :param chunk_size: prompts per generate() call during rollouts
    :param init_kl_coef: initial KL penalty coefficient
    :param target: adaptive-KL target (None -> fixed coefficient)
    :param horizon: adaptive-KL horizon
    :param gamma, lam: GAE discount / lambda
    :param cliprange, cliprange_value: PPO ratio / value clipping
    :param vf_coef: value loss coefficient
    :param scale_reward: None | "ref" | "running" reward scaling
    :param cliprange_reward: reward clip bound
    :param gen_kwargs: generation settings for rollouts/eval
    :param gen_experience_kwargs: override generation settings for experience
    """

    name: str = "PPOConfig"
    ppo_epochs: int = 4
    num_rollouts: int = 128
    chunk_size: int = 128
    init_kl_coef: float = 0.001
    target: Optional[float] = None
    horizon: int = 10000
    gamma: float = 1.0
    lam: float = 0.95
    cliprange: float = 0.2
    cliprange_value: float = 0.2
    vf_coef: float = 1.0
    scale_reward: Optional[str] = "ignored"
    ref_mean: Optional[float] = None
    ref_std: Optional[float] = None
    cliprange_reward: float = 10.0
    num_value_layers_unfrozen: int = 0
    # score rollouts on EVERY rank instead of the reference's gather-to-rank-0
    # + scatter protocol (accelerate_ppo_trainer.py:292-338).  Correct only
    # for stateless/deterministic reward fns; the NeMo path always scores
    # locally (nemo_ppo_trainer.py:195-197).  At DP=8 the rank-0 round trip
    # serializes ~10 ms per chunk while 7 ranks idle.
    local_rewards: bool = False
    gen_kwargs: Dict[str, Any] = field(default_factory=lambda: dict(max_new_tokens=40, top_k=0, top_p=1.0, do_sample=True))
    gen_experience_kwargs: Optional[Dict[str, Any]] = None

    def get_advantages_and_returns(
        self,
        values: torch.Tensor,
        rewards: torch.Tensor,
        response_length: int,
        use_whitening: bool = True,
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """GAE over the response tokens (reference modeling_ppo.py:136-173);
        the reverse scan is the affine-scan HIP kernel on GPU."""
        return ops.gae_advantages_and_returns(
            values[:, :response_length].float(),
            rewards[:, :response_length].float(),
            self.gamma,
            self.lam,
            use_whitening=use_whitening,
        )

    def loss(
        self,
        logprobs: torch.Tensor,
        values: torch.Tensor,
        old_logprobs: torch.Tensor,
        old_values: torch.Tensor,
        advantages: torch.Tensor,
        returns: torch.Tensor,
        mask: torch.Tensor,
    ):
        """Clipped surrogate PPO loss (reference modeling_ppo.py:175-238)."""
        values_clipped = torch.clamp(
            values, old_values - self.cliprange_value, old_values + self.cliprange_value
        )
        n = mask.sum()

        vf_loss1 = (values - returns) ** 2
        vf_loss2 = (values_clipped - returns) ** 2
        vf_loss = 0.5 * torch.sum(torch.max(vf_loss1, vf_loss2) * mask) / n
        vf_clipfrac = torch.sum((vf_loss2 > vf_loss1).float() * mask) / n

        log_ratio = (logprobs - old_logprobs) * mask
        ratio = torch.exp(log_ratio)
        # k3 KL estimator
        with torch.no_grad():
            approx_kl = torch.mean((ratio - 1) - log_ratio)

        pg_loss1 = -advantages * ratio
        pg_loss2 = -advantages * torch.clamp(ratio, 1.0 - self.cliprange, 1.0 + self.cliprange)
        pg_loss = torch.sum(torch.max(pg_loss1, pg_loss2) * mask) / n
        pg_clipfrac = torch.sum((pg_loss2 > pg_loss1).float() * mask) / n

        loss = pg_loss + self.vf_coef * vf_loss

        # stats stay DEVICE TENSORS (no .item() sync in the hot loop; the
        # tracker converts at logging time)
        stats = dict(
            losses=dict(
                total_loss=loss.detach(),
                policy_loss=pg_loss.detach(),
                value_loss=vf_loss.detach(),
            ),
            values=dict(
                get_tensor_stats(values.detach(), mask, n),
                values_error=torch.sum(((values.detach() - returns) * mask) ** 2) / n,
                clipfrac=vf_clipfrac,
            ),
            old_values=get_tensor_stats(old_values, mask, n),
            returns=get_tensor_stats(returns, mask, n),
            policy=dict(approx_kl=approx_kl, clipfrac=pg_clipfrac),
            ratio=(ratio.detach() * mask).sum() / n,
            padding_percentage=1 - n / mask.numel(),
        )
        return loss, flatten_dict(stats)


@dataclass
class CausalLMOutputWithValue:
    logits: Optional[torch.Tensor] = None
    values: Optional[torch.Tensor] = None
    ref_logits: Optional[torch.Tensor] = None
    last_hidden_state: Optional[torch.Tensor] = None


class AutoModelForCausalLMWithValueHead(PreTrainedModelWrapper):
    """Native LM + scalar value head (reference modeling_ppo.py:266-382)."""

    _supported_modules = ["v_head", "v_branch"]
    _supported_args = ["peft_config", "num_value_layers_unfrozen"]

    def __init__(self, base_model: CausalTransformer, peft_config=None, num_value_layers_unfrozen: int = 0):
        super().__init__(base_model)
        self.peft_config = peft_config
        self.num_value_layers_unfrozen = num_value_layers_unfrozen
        # value head computes in fp32 regardless of trunk dtype
        self.v_head = make_head(self.config.hidden_size, 1, dtype=torch.float32)
        # num_value_layers_unfrozen > 0: the value function gets its own
        # trainable copies of the top layers (reference make_value_branch)
        self.v_branch = None
        if num_value_layers_unfrozen > 0:
            self.v_branch = ValueBranch(base_model, num_value_layers_unfrozen)

    def _value_stash(self):
        return [-self.num_value_layers_unfrozen] if self.v_branch is not None else []

    def _branch_values(self, out, input_ids, attention_mask, position_ids, logits_slice):
        hidden = out.hidden_at_layer
        if isinstance(hidden, dict):
            hidden = hidden[-self.num_value_layers_unfrozen]
        ctx = self.base_model.make_context(input_ids, attention_mask, 0)
        if position_ids is not None:
            ctx.position_ids = position_ids.to(torch.int32)
        return self.v_branch(hidden, ctx, self.base_model.rope_tables, logits_slice)

    def forward(
        self,
        input_ids: torch.Tensor,
        attention_mask: Optional[torch.Tensor] = None,
        position_ids: Optional[torch.Tensor] = None,
        return_ref_logits: bool = False,
        logits_slice=None,
        return_logits: bool = True,
        **kwargs,
    ) -> CausalLMOutputWithValue:
        out = self.base_model(
            input_ids, attention_mask=attention_mask, position_ids=position_ids,
            logits_slice=logits_slice, hidden_at_layer=self._value_stash() or None,
            return_logits=return_logits,
        )
        if self.v_branch is not None:
            values = self._branch_values(out, input_ids, attention_mask, position_ids,
                                         logits_slice)
        else:
            hs = out.last_hidden_state
            if logits_slice is not None:
                hs = hs[:, logits_slice[0] : logits_slice[1]].contiguous()
            values = self.v_head(hs.to(self.v_head[0].weight.dtype)).squeeze(-1).float()
        return CausalLMOutputWithValue(
            logits=out.logits, values=values, last_hidden_state=out.last_hidden_state
        )

    def generate(self, input_ids, attention_mask=None, **kwargs):
        return generate(self.base_model, input_ids, attention_mask, **kwargs)

    def generate_eval(self, input_ids, attention_mask=None, **kwargs):
        return self.generate(input_ids, attention_mask, **kwargs)

    @torch.no_grad()
    def forward_experience(self, input_ids, attention_mask, lo: int, hi: int,
                           labels: torch.Tensor):
        """Fused experience path without a frozen branch (ref handled by the
        caller); see the hydra variant for semantics."""
        out = self.base_model(input_ids, attention_mask=attention_mask, return_logits=False,
                              hidden_at_layer=self._value_stash() or None)
        h = out.last_hidden_state[:, lo:hi].contiguous()
        B, T = h.shape[:2]
        if self.v_branch is not None:
            values = self._branch_values(out, input_ids, attention_mask, None, (lo, hi))
        else:
            values = self.v_head(h.to(self.v_head[0].weight.dtype)).squeeze(-1).float()
        lm = self.base_model.lm_head
        flat_labels = labels.reshape(-1)
        import os

        if (os.environ.get("TRLX_AMD_FUSED_LM_LOGPROBS") != "0"
                and h.is_cuda and h.dtype == torch.bfloat16 and lm.bias is None
                and h.shape[-1] % 32 == 0):
            logprobs = ops.lm_logprobs(h.reshape(-1, h.shape[-1]).contiguous(), lm.weight,
                                       flat_labels).view(B, T)
        else:
            logprobs = ops.logprobs_of_labels(lm(h), labels)
        return logprobs, None, values


class ValueBranch(nn.Module):
    """Trainable value branch: copies of the top ``num_value_layers_unfrozen``
    blocks + final norm + scalar head, fed from the stashed pre-branch hidden
    state — the value function gets its own layers decoupled from the policy
    (reference modeling_ppo.py:255-263 ``make_value_branch``)."""

    def __init__(self, base_model: CausalTransformer, num_layers: int):
        super().__init__()
        self.num_layers = num_layers
        self.blocks = nn.ModuleList(
            copy.deepcopy(block) for block in base_model.layers[-num_layers:]
        )
        self.final_norm = copy.deepcopy(base_model.final_norm)
        self.v_head = make_head(base_model.config.hidden_size, 1, dtype=torch.float32)
        for p in self.parameters():  # copies of frozen trunk layers train here
            p.requires_grad_(True)

    def forward(self, hidden: torch.Tensor, ctx, rope_tables, logits_slice=None) -> torch.Tensor:
        h, res = hidden, None
        for block in self.blocks:
            h, res = block(h, ctx, rope_tables, res=res)
        if logits_slice is not None:
            h = h[:, logits_slice[0] : logits_slice[1]].contiguous()
            if res is not None:
                res = res[:, logits_slice[0] : logits_slice[1]].contiguous()
        if res is None:
            h = self.final_norm(h)
        else:
            h = self.final_norm.forward_add(h, res)[0]
        return self.v_head(h.to(self.v_head[0].weight.dtype)).squeeze(-1).float()


class FrozenBranch(nn.Module):
    """Frozen copies of the top ``num_layers_unfrozen`` blocks + final norm +
    lm_head — the hydra reference head (reference ModelBranch,
    modeling_ppo.py:502-544), arch-agnostic here."""

    def __init__(self, base_model: CausalTransformer, num_layers_unfrozen: int):
        super().__init__()
        self.num_layers_unfrozen = num_layers_unfrozen
        self.blocks = nn.ModuleList(
            copy.deepcopy(block) for block in base_model.layers[-num_layers_unfrozen:]
        )
        self.final_norm = copy.deepcopy(base_model.final_norm)
        self.lm_head = copy.deepcopy(base_model.lm_head)
        for p in self.parameters():
            p.requires_grad_(False)

    def forward(self, hidden: torch.Tensor, ctx, rope_tables, logits_slice=None) -> torch.Tensor:
        with torch.no_grad():
            h = self.forward_hidden(hidden, ctx, rope_tables, logits_slice)
            return self.lm_head(h)

    def forward_hidden(self, hidden: torch.Tensor, ctx, rope_tables, logits_slice=None) -> torch.Tensor:
        """Final-norm hidden states of the frozen branch (pre-lm_head) — the
        fused lm_logprobs path consumes these directly."""
        with torch.no_grad():
            h, res = hidden, None
            if self._offloaded and h.is_cuda:
                h, res = self._forward_blocks_offloaded(h, ctx, rope_tables)
            else:
                for block in self.blocks:
                    h, res = block(h, ctx, rope_tables, res=res)
            if logits_slice is not None:
                h = h[:, logits_slice[0] : logits_slice[1]]
                if res is not None:
                    res = res[:, logits_slice[0] : logits_slice[1]]
            if res is None:
                return self.final_norm(h)
            return self.final_norm.forward_add(h, res)[0]

    # ---- K15: reference-weight CPU offload ---------------------------------
    # (reference modeling_nemo_ppo.py:228-244 offload_reference_model for the
    # 65B config, megatron_65b.yaml:5-6).  On MI355X the 288 GB HBM keeps
    # <=20B hydras resident, so this is OPT-IN (model.ref_offload) for
    # 65B-class replicas: block weights live in pinned host memory and stream
    # to two GPU shadow blocks just in time, the H2D of block i+1 overlapping
    # block i's compute on a side stream.

    _offloaded = False

    def offload(self):
        """Move the frozen blocks' weights to pinned host memory; forwards
        stream them back block-by-block (double-buffered).  final_norm and
        lm_head stay resident (tiny next to the blocks)."""
        if self._offloaded or len(self.blocks) == 0:
            return self
        dev = next(self.blocks[0].parameters()).device
        if dev.type == "cuda":
            # two GPU shadow blocks, structure-identical to a frozen block
            self._shadows = [copy.deepcopy(self.blocks[0]), copy.deepcopy(self.blocks[0])]
            self._copy_stream = torch.cuda.Stream()
            self._ready = [torch.cuda.Event(), torch.cuda.Event()]
            self._free = [torch.cuda.Event(), torch.cuda.Event()]
        pin = torch.cuda.is_available()
        for block in self.blocks:
            for p in list(block.parameters()) + list(block.buffers()):
                host = p.data.detach().cpu()
                p.data = host.pin_memory() if pin else host
        self._offloaded = True
        return self

    def _forward_blocks_offloaded(self, h, ctx, rope_tables):
        shadows, ready, free = self._shadows, self._ready, self._free
        cs = self._copy_stream
        main = torch.cuda.current_stream()

        def load(slot, block):
            with torch.cuda.stream(cs):
                cs.wait_event(free[slot])
                for dst, src in zip(
                    list(shadows[slot].parameters()) + list(shadows[slot].buffers()),
                    list(block.parameters()) + list(block.buffers()),
                ):
                    dst.data.copy_(src.data, non_blocking=True)
                ready[slot].record(cs)

        # both shadows start free (record on main so the first waits are no-ops)
        free[0].record(main)
        free[1].record(main)
        load(0, self.blocks[0])
        res = None
        for i, block in enumerate(self.blocks):
            slot = i % 2
            if i + 1 < len(self.blocks):
                load((i + 1) % 2, self.blocks[i + 1])
            main.wait_event(ready[slot])
            h, res = shadows[slot](h, ctx, rope_tables, res=res)
            free[slot].record(main)  # shadow reusable once this compute drains
        return h, res


class AutoModelForCausalLMWithHydraValueHead(AutoModelForCausalLMWithValueHead):
    """Value-head LM + frozen reference branch sharing the trunk
    (reference modeling_ppo.py:385-499)."""

    _supported_modules = ["v_head", "frozen_head"]
    _supported_args = ["peft_config", "num_layers_unfrozen", "num_value_layers_unfrozen"]

    def __init__(self, base_model: CausalTransformer, peft_config=None,
                 num_layers_unfrozen: int = -1, num_value_layers_unfrozen: int = 0):
        super().__init__(base_model, peft_config=peft_config,
                         num_value_layers_unfrozen=num_value_layers_unfrozen)
        self.num_layers_unfrozen = num_layers_unfrozen
        self.frozen_head = None
        if num_layers_unfrozen > 0 and peft_config is None:
            self.frozen_head = FrozenBranch(base_model, num_layers_unfrozen)

    def forward(
        self,
        input_ids: torch.Tensor,
        attention_mask: Optional[torch.Tensor] = None,
        position_ids: Optional[torch.Tensor] = None,
        return_ref_logits: bool = False,
        logits_slice=None,
        return_logits: bool = True,
        **kwargs,
    ) -> CausalLMOutputWithValue:
        """One trunk pass; optionally also the reference logits via the frozen
        branch on the stashed pre-branch hidden state (MI355X fusion of the
        reference's forward + forward_hydra pair)."""
        stash = list(self._value_stash())
        if return_ref_logits and self.frozen_head is not None:
            stash.append(-self.num_layers_unfrozen)
        out = self.base_model(
            input_ids, attention_mask=attention_mask, position_ids=position_ids,
            hidden_at_layer=stash or None, logits_slice=logits_slice,
            return_logits=return_logits,
        )
        if self.v_branch is not None:
            values = self._branch_values(out, input_ids, attention_mask, position_ids,
                                         logits_slice)
        else:
            hs = out.last_hidden_state
            if logits_slice is not None:
                hs = hs[:, logits_slice[0] : logits_slice[1]].contiguous()
            values = self.v_head(hs.to(self.v_head[0].weight.dtype)).squeeze(-1).float()
        ref_logits = None
        if return_ref_logits and self.frozen_head is not None:
            ctx = self.base_model.make_context(input_ids, attention_mask, 0)
            if position_ids is not None:
                ctx.position_ids = position_ids.to(torch.int32)
            ref_hidden = out.hidden_at_layer
            if isinstance(ref_hidden, dict):
                ref_hidden = ref_hidden[-self.num_layers_unfrozen]
            ref_logits = self.frozen_head(ref_hidden, ctx, self.base_model.rope_tables,
                                          logits_slice=logits_slice)
        elif return_ref_logits and self.peft_config is not None:
            # peft hydra: the base model WITHOUT adapters is the reference
            # (reference accelerate_ppo_trainer.py:74-77 + peft disable_adapter)
            from .lora import adapters_disabled

            with torch.no_grad(), adapters_disabled(self.base_model):
                ref_logits = self.base_model(
                    input_ids, attention_mask=attention_mask, position_ids=position_ids,
                    logits_slice=logits_slice,
                ).logits
        return CausalLMOutputWithValue(
            logits=out.logits, values=values, ref_logits=ref_logits,
            last_hidden_state=out.last_hidden_state,
        )

    @torch.no_grad()
    def forward_experience(self, input_ids, attention_mask, lo: int, hi: int,
                           labels: torch.Tensor):
        """Experience-phase fused path: per-token logprobs (policy + frozen
        reference) and values for positions [lo, hi) in ONE trunk pass, using
        the hand-MFMA lm_logprobs kernel — the [B, T, V] logits never exist.
        ``labels`` = input_ids[:, lo+1 : hi+1].  Returns
        (logprobs, ref_logprobs, values) each [B, hi-lo]; ref is None when
        there is no frozen branch (caller falls back to a separate ref model
        or adapter toggling)."""
        stash = list(self._value_stash())
        if self.frozen_head is not None:
            stash.append(-self.num_layers_unfrozen)
        out = self.base_model(input_ids, attention_mask=attention_mask,
                              hidden_at_layer=stash or None, return_logits=False)
        h = out.last_hidden_state[:, lo:hi].contiguous()
        B, T = h.shape[:2]
        if self.v_branch is not None:
            values = self._branch_values(out, input_ids, attention_mask, None, (lo, hi))
        else:
            values = self.v_head(h.to(self.v_head[0].weight.dtype)).squeeze(-1).float()
        lm = self.base_model.lm_head

        def fused_ok(head):
            # default ON: the pipelined lm_logprobs_v2 kernel beats the
            # hipBLASLt GEMM + logprob-gather path 1.34-1.41x on the
            # experience shapes ([1312-5248, 50257, 768]: 157/273/512 us vs
            # 221/385/685 us, exact numerics).  Set
            # TRLX_AMD_FUSED_LM_LOGPROBS=0 to fall back.
            import os

            return (os.environ.get("TRLX_AMD_FUSED_LM_LOGPROBS") != "0"
                    and h.is_cuda and h.dtype == torch.bfloat16 and head.bias is None
                    and h.shape[-1] % 32 == 0)

        flat_labels = labels.reshape(-1)
        if fused_ok(lm):
            logprobs = ops.lm_logprobs(h.reshape(-1, h.shape[-1]).contiguous(), lm.weight,
                                       flat_labels).view(B, T)
        else:
            logprobs = ops.logprobs_of_labels(lm(h), labels)

        ref_logprobs = None
        if self.frozen_head is not None:
            ctx = self.base_model.make_context(input_ids, attention_mask, 0)
            ref_hidden = out.hidden_at_layer
            if isinstance(ref_hidden, dict):
                ref_hidden = ref_hidden[-self.num_layers_unfrozen]
            rh = self.frozen_head.forward_hidden(ref_hidden, ctx,
                                                 self.base_model.rope_tables,
                                                 logits_slice=(lo, hi))
            rlm = self.frozen_head.lm_head
            if fused_ok(rlm):
                ref_logprobs = ops.lm_logprobs(rh.reshape(-1, rh.shape[-1]).contiguous(),
                                               rlm.weight, flat_labels).view(B, T)
            else:
                ref_logprobs = ops.logprobs_of_labels(rlm(rh), labels)
        elif self.peft_config is not None:
            from .lora import adapters_disabled

            with adapters_disabled(self.base_model):
                rout = self.base_model(input_ids, attention_mask=attention_mask,
                                       return_logits=False)
            rh = rout.last_hidden_state[:, lo:hi].contiguous()
            if fused_ok(lm):
                ref_logprobs = ops.lm_logprobs(rh.reshape(-1, rh.shape[-1]).contiguous(),
                                               lm.weight, flat_labels).view(B, T)
            else:
                ref_logprobs = ops.logprobs_of_labels(lm(rh), labels)
        return logprobs, ref_logprobs, values

    def forward_hydra(
        self,
        input_ids: torch.Tensor,
        attention_mask: Optional[torch.Tensor] = None,
        position_ids: Optional[torch.Tensor] = None,
        **kwargs,
    ) -> CausalLMOutputWithValue:
        """Reference-branch-only forward (parity with reference
        modeling_ppo.py:410-453); prefer forward(return_ref_logits=True)."""
        if self.frozen_head is None:
            raise RuntimeError("forward_hydra requires num_layers_unfrozen > 0")
        out = self.base_model(
            input_ids, attention_mask=attention_mask, position_ids=position_ids,
            hidden_at_layer=-self.num_layers_unfrozen, return_logits=False,
        )
        ctx = self.base_model.make_context(input_ids, attention_mask, 0)
        if position_ids is not None:
            ctx.position_ids = position_ids.to(torch.int32)
        ref_logits = self.frozen_head(out.hidden_at_layer, ctx, self.base_model.rope_tables)
        return CausalLMOutputWithValue(logits=ref_logits)
