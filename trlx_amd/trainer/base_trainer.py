"""The native base trainer — owner of the training loop and the runtime.

Parity target: reference trlx/trainer/accelerate_base_trainer.py
(AccelerateRLTrainer), with the accelerate/DeepSpeed machinery replaced by the
native runtime: torch.distributed over RCCL (one process per GPU), the arena
FusedAdamW (parallel/optim.py), the bucketed GradReducer (parallel/ddp.py),
bf16 model weights with fp32 masters, and jsonl/tensorboard trackers.
"""

import contextlib
import json
import os
import sys
from abc import abstractmethod
from time import time
from typing import Dict, List, Optional, Tuple

import torch

from ..data.configs import TRLConfig
from ..parallel import comm
from ..parallel.ddp import GradReducer
from ..parallel.optim import FusedAdamW, build_optimizer
from ..pipeline import MiniBatchIterator
from ..trainer import BaseRLTrainer, register_trainer
from ..utils import (
    Clock,
    filter_non_scalars,
    get_distributed_config,
    get_git_tag,
    get_scheduler_class,
    significant,
)
from ..utils import logging
from ..utils.modeling import (
    flatten_dict,
    freeze_bottom_causal_layers,
    freeze_bottom_seq2seq_layers,
    gather_dict,
)
from ..utils.tokenizer import get_tokenizer
from ..utils.trackers import make_tracker

logger = logging.get_logger(__name__)

# live trainers (weak) for deterministic hipGraph teardown — see
# models/nn/generation.release_graphs
import weakref  # noqa: E402

_TRAINERS = weakref.WeakSet()


def release_train_graphs():
    """Destroy all captured train-step graphs now (ordered, synchronized)."""
    if not torch.cuda.is_available():
        return
    torch.cuda.synchronize()
    for trainer in list(_TRAINERS):
        graphs = getattr(trainer, "_train_graphs", None)
        if not graphs:
            continue
        for key, entry in list(graphs.items()):
            if entry and entry is not False:
                try:
                    entry[0].reset()
                except (RuntimeError, AttributeError):
                    pass
        trainer._train_graphs = {}
    torch.cuda.synchronize()


@register_trainer
class NativeRLTrainer(BaseRLTrainer):
    """Runtime + training loop shared by PPO/ILQL/SFT/RFT trainers."""

    def __init__(self, config: TRLConfig, **kwargs):
        super().__init__(config, **kwargs)

        comm.init_distributed()
        from ..parallel import topo

        if (config.train.pipeline_parallel_size > 1
                and config.train.sequence_parallel):
            raise NotImplementedError(
                "sequence_parallel + pipeline_parallel_size > 1 is not supported yet; "
                "disable one of them."
            )
        topo.init_model_parallel(config.train.tensor_parallel_size,
                                 config.train.pipeline_parallel_size)
        self.device = comm.get_device()
        self.world_size = comm.world_size()
        self.dp_size = topo.dp_size()
        self.dp_group = topo.dp_group()
        self.tp_size = topo.tp_size()
        self.pp_size = topo.pp_size()
        self.local_rank = comm.local_rank()
        # re-seed with the DP rank so TP peers share RNG streams (identical
        # rollouts within a TP group, decorrelated across DP replicas —
        # reference modeling_nemo_ppo.py:384-393)
        from ..utils import set_seed

        set_seed(config.train.seed, rank_offset=topo.dp_rank())
        comm.barrier()

        self.mb_count = 0
        self.max_length = config.train.seq_length
        if config.train.minibatch_size:
            assert config.train.batch_size % config.train.minibatch_size == 0, \
                "Minibatch size must divide batch size"
            self.mb_size = config.train.minibatch_size
        else:
            self.mb_size = config.train.batch_size
        self.num_mb = config.train.batch_size // self.mb_size

        # dtype policy: bf16 weights on GPU (fp32 masters live in the
        # optimizer); fp32 on CPU
        if self.device.type == "cuda" and config.train.mixed_precision == "bf16":
            self.dtype = torch.bfloat16
        else:
            self.dtype = torch.float32

        self.tokenizer = get_tokenizer(
            config.tokenizer.tokenizer_path,
            padding_side=config.tokenizer.padding_side,
            truncation_side=config.tokenizer.truncation_side,
            **config.tokenizer.tokenizer_extra_configs,
        )

        self.model = self.setup_model()
        self.opt = self.setup_optimizer()
        self.scheduler = self.setup_scheduler()
        self.reducer = GradReducer(
            self.opt, self.model, bucket_size_mb=config.train.bucket_size_mb,
            average=not isinstance(self.opt, FusedAdamW),
            zero=config.train.zero_stage >= 1 and isinstance(self.opt, FusedAdamW),
            process_group=self.dp_group,
        )
        self.reducer.broadcast_parameters(self.model)
        if (getattr(self.config.model, "ref_offload", False)
                and getattr(self.model, "frozen_head", None) is not None):
            # K15 ref-weight CPU offload — AFTER the init broadcast (RCCL
            # can't broadcast the pinned-host tensors this creates)
            self.model.frozen_head.offload()

        script_name = os.path.basename(sys.argv[0]).rsplit(".", 1)[0]
        run_time = int(time())
        if config.train.run_name:
            self.run_name = config.train.run_name
        else:
            self.run_name = f"{script_name}/{run_time}"

        logging_dir = config.train.logging_dir or os.path.join("runs", self.run_name.replace("/", "_"))
        run_config = {
            **config.to_dict(),
            "distributed": get_distributed_config(),
            "git": get_git_tag(),
        }
        self.tracker = make_tracker(
            config.train.tracker, logging_dir, run_config, main_process=comm.is_main_process()
        )

        self.generate_kwargs = dict(getattr(config.method, "gen_kwargs", {}))
        gen_exp = getattr(config.method, "gen_experience_kwargs", None)
        self.generate_experience_kwargs = dict(gen_exp) if gen_exp else None
        # gen-kwarg sweeps: a list-valued generate kwarg becomes an eval sweep
        self.generate_sweep_kwarg = None
        for k, v in self.generate_kwargs.items():
            if isinstance(v, list):
                if self.generate_sweep_kwarg is not None:
                    logger.info("Only a single sweep is allowed, {k} is going to be set to its first value")
                    self.generate_kwargs[k] = v[0]
                else:
                    self.generate_sweep_kwarg = (k, v)
        if self.generate_sweep_kwarg is not None:
            self.generate_kwargs.pop(self.generate_sweep_kwarg[0])

        self.iter_count = 0
        self.nth_evaluation = 0
        _TRAINERS.add(self)

    # --- setup ---------------------------------------------------------------

    def setup_model(self):
        model = self.get_arch(self.config)
        if getattr(model, "is_pipelined", False):
            # PP model owns its stage-aware freezing; SP/checkpointing toggles
            # are guarded at config validation
            model.freeze_bottom(self.config.model.num_layers_unfrozen)
            model = model.to(self.device)
            if self.dtype != torch.float32:
                model.cast_compute(self.dtype)
            return model
        base = model.base_model if hasattr(model, "base_model") else model
        if self.config.model.peft_config is not None:
            pass  # LoRA already froze the base
        elif self.config.model.model_arch_type == "seq2seq":
            freeze_bottom_seq2seq_layers(base, self.config.model.num_layers_unfrozen)
        else:
            freeze_bottom_causal_layers(base, self.config.model.num_layers_unfrozen)
        if (self.config.train.sequence_parallel and self.tp_size > 1
                and hasattr(base, "set_sequence_parallel")):
            base.set_sequence_parallel(True)
        if getattr(self.config.model, "gradient_checkpointing", False):
            base.gradient_checkpointing = True
        model = model.to(self.device)
        if self.dtype != torch.float32:
            # heads stay fp32; trunk + frozen branch go bf16
            if hasattr(model, "cast_compute"):
                model.cast_compute(self.dtype)
            else:
                base.to(self.dtype)
        return model

    def setup_optimizer(self):
        from ..parallel import topo

        return build_optimizer(
            self.model, self.config.optimizer.name, self.config.optimizer.kwargs,
            world=topo.dp_size(), zero=self.config.train.zero_stage >= 1,
        )

    def setup_scheduler(self):
        from ..parallel.optim import NullOptimizer, NullScheduler

        if isinstance(self.opt, NullOptimizer):
            return NullScheduler()
        cls = get_scheduler_class(self.config.scheduler.name)
        return cls(self.opt, **self.config.scheduler.kwargs)

    @property
    def unwrapped_model(self):
        return self.model

    # --- decoding / generation ----------------------------------------------

    def decode(self, prompts, samples, prompt_sizes=None, append_eos_token: bool = False
               ) -> Tuple[List[str], List[str], List[str]]:
        """Token tensors -> (samples, prompts, outputs) strings with
        stop-sequence trimming (reference accelerate_base_trainer.py:203-254)."""
        if prompt_sizes is None:
            prompt_sizes = [len(prompts[0])] * len(prompts)

        # ONE device->host transfer per batch: decoding row-by-row from GPU
        # tensors costs a ~50us sync per row (measured: 15 ms of a 190 ms
        # PPO cycle at chunk=128)
        def to_lists(x):
            if torch.is_tensor(x):
                return x.cpu().tolist()
            if isinstance(x, (list, tuple)) and len(x) and torch.is_tensor(x[0]):
                return [t.cpu().tolist() for t in x]
            return x

        prompts = to_lists(prompts)
        samples = to_lists(samples)
        prompt_sizes = to_lists(prompt_sizes)

        str_samples, str_prompts, str_outputs = [], [], []
        for prompt, sample, prompt_size in zip(prompts, samples, prompt_sizes):
            if self.config.model.model_arch_type == "seq2seq":
                output_start_ix = 0
            else:
                output_start_ix = prompt_size
            str_prompt = self.tokenizer.decode(prompt[:prompt_size], skip_special_tokens=True)
            str_output = self.tokenizer.decode(sample[output_start_ix:], skip_special_tokens=True)
            trimmed = False
            if self.stop_sequences:
                for stop in self.stop_sequences:
                    stop_ix = str_output.find(stop)
                    if stop_ix >= 0:
                        str_output = str_output[:stop_ix].rstrip()
                        trimmed = True
            last = sample[-1] if len(sample) else -1
            if append_eos_token and (
                trimmed or last == self.tokenizer.eos_token_id or last == self.tokenizer.pad_token_id
            ):
                str_output += self.tokenizer.eos_token

            str_prompts.append(str_prompt)
            str_outputs.append(str_output)
            if self.config.model.model_arch_type == "seq2seq":
                str_samples.append(str_prompt + self.tokenizer.sep_token + str_output)
            else:
                str_samples.append(str_prompt + str_output)
        return str_samples, str_prompts, str_outputs

    def generate(self, input_ids, attention_mask=None, **kwargs):
        """Rollout generation with the method's experience kwargs."""
        input_ids = input_ids.to(self.device)
        if attention_mask is not None:
            attention_mask = attention_mask.to(self.device)
        if self.generate_experience_kwargs is not None:
            kwargs = dict(self.generate_experience_kwargs, **kwargs)
        else:
            kwargs = dict(self.generate_kwargs, **kwargs)
        kwargs.setdefault("eos_token_id", self.tokenizer.eos_token_id)
        kwargs.setdefault("pad_token_id", self.tokenizer.pad_token_id)
        with torch.no_grad():
            return self.unwrapped_model.generate(input_ids, attention_mask=attention_mask, **kwargs)

    def generate_eval(self, input_ids, attention_mask=None, **kwargs):
        input_ids = input_ids.to(self.device)
        if attention_mask is not None:
            attention_mask = attention_mask.to(self.device)
        kwargs = dict(self.generate_kwargs, **kwargs)
        kwargs.setdefault("eos_token_id", self.tokenizer.eos_token_id)
        kwargs.setdefault("pad_token_id", self.tokenizer.pad_token_id)
        with torch.no_grad():
            return self.unwrapped_model.generate(input_ids, attention_mask=attention_mask, **kwargs)

    # --- persistence ----------------------------------------------------------

    def save_pretrained(self, directory: Optional[str] = None, **kwargs):
        from ..parallel import topo

        if directory is None:
            directory = os.path.join(self.config.train.checkpoint_dir, "hf_model")
        comm.barrier()
        if self.tp_size > 1 or self.pp_size > 1:
            # sharded formats: every model-parallel rank of DP replica 0
            # writes its own mp_rank_XX[_YYY] shard
            if topo.dp_rank() == 0:
                self.unwrapped_model.save_pretrained(directory)
            if comm.is_main_process() and self.tokenizer:
                self.tokenizer.save_pretrained(directory)
        elif comm.is_main_process():
            self.unwrapped_model.save_pretrained(directory)
            if self.tokenizer:
                self.tokenizer.save_pretrained(directory)
        comm.barrier()

    def _state_file(self) -> str:
        """Per-model-parallel-rank training-state filename (plain ``state.pt``
        when there is no model parallelism)."""
        from ..parallel import topo

        if self.tp_size > 1 or self.pp_size > 1:
            return f"state_mp_{topo.tp_rank():02d}_{topo.pp_rank():03d}.pt"
        return "state.pt"

    def save(self, directory: Optional[str] = None, **kwargs):
        """Training-state checkpoint: model + optimizer + scheduler + step.
        Under TP/PP every model-parallel rank of DP replica 0 writes its own
        state shard."""
        from ..parallel import topo

        dst_dir = directory or self.config.train.checkpoint_dir
        comm.barrier()
        if topo.dp_rank() == 0:
            os.makedirs(dst_dir, exist_ok=True)
            state = {
                "model": self.unwrapped_model.state_dict(),
                "scheduler": self.scheduler.state_dict(),
                "iter_count": self.iter_count,
            }
            if self.config.train.save_optimizer:
                state["optimizer"] = self.opt.state_dict()
            torch.save(state, os.path.join(dst_dir, self._state_file()))
            if comm.is_main_process():
                with open(os.path.join(dst_dir, "state.json"), "w") as f:
                    json.dump({"iter_count": self.iter_count}, f)
        comm.barrier()

    def load(self, directory: Optional[str] = None, **kwargs):
        src_dir = directory or self.config.train.checkpoint_dir
        path = os.path.join(src_dir, self._state_file())
        state = torch.load(path, map_location=self.device, weights_only=True)
        self.unwrapped_model.load_state_dict(state["model"])
        if "optimizer" in state and self.config.train.save_optimizer:
            self.opt.load_state_dict(state["optimizer"])
        self.scheduler.load_state_dict(state["scheduler"])
        self.iter_count = state.get("iter_count", 0)

    # --- evaluation ------------------------------------------------------------

    def evaluate(self) -> Dict:
        """Generate on eval prompts, score with reward_fn/metric_fn on rank 0
        (reference accelerate_base_trainer.py:339-500)."""
        logger.info("Evaluating model")
        if self.generate_sweep_kwarg is not None:
            gen_sweep_arg, gen_sweep_values = self.generate_sweep_kwarg
        else:
            gen_sweep_arg, gen_sweep_values = None, [None]

        stats = {}
        table = []
        columns = ["prompt", "output"]

        for i_sweep, gen_sweep_value in enumerate(gen_sweep_values):
            sweep_suffix = f"@{gen_sweep_arg}={gen_sweep_value}" if gen_sweep_value is not None else ""

            all_samples, all_prompts, all_prompt_sizes, all_metadata = [], [], [], []
            generate_time = time()
            for prompts in self.eval_dataloader:
                metadata = {k: v for k, v in prompts.items() if k not in ("input_ids", "attention_mask")}
                gen_kwargs = {gen_sweep_arg: gen_sweep_value} if gen_sweep_value is not None else {}
                samples = self.generate_eval(prompts["input_ids"], prompts["attention_mask"], **gen_kwargs)
                if self.config.model.model_arch_type == "seq2seq":
                    samples = samples[:, 1:].contiguous()

                prompt_sizes = torch.full((len(prompts["input_ids"]),), prompts["input_ids"].shape[1],
                                          device=samples.device, dtype=torch.long)
                pad_id = self.tokenizer.pad_token_id
                prompts_t = comm.pad_across_processes(prompts["input_ids"].to(samples.device), 1, pad_id)
                samples_t = comm.pad_across_processes(samples, 1, pad_id)
                all_samples.extend(comm.gather(samples_t).tolist())
                all_prompts.extend(comm.gather(prompts_t).tolist())
                all_prompt_sizes.extend(comm.gather(prompt_sizes).tolist())
                all_metadata.append(gather_dict(metadata))

            stats["time/generate"] = time() - generate_time

            if comm.is_main_process():
                str_samples, str_prompts, str_outputs = self.decode(all_prompts, all_samples, all_prompt_sizes)
                columns = ["prompt", "output"]
                columns_data = [str_prompts, str_outputs]

                metadata = {}
                if all_metadata:
                    metadata, *xs = all_metadata
                    for k in metadata:
                        for x in xs:
                            metadata[k].extend(x[k])

                if self.reward_fn:
                    rewards = self.reward_fn(
                        samples=str_samples, prompts=str_prompts, outputs=str_outputs,
                        tokenizer=self.tokenizer, **metadata,
                    )
                    if isinstance(rewards, torch.Tensor):
                        rewards = rewards.detach().float().cpu().view(len(str_samples), -1).sum(-1)
                    elif len(rewards) and isinstance(rewards[0], torch.Tensor):
                        rewards = torch.tensor([r.sum().item() for r in rewards], dtype=torch.float)
                    elif len(rewards) and isinstance(rewards[0], list):
                        rewards = torch.tensor([sum(r) for r in rewards], dtype=torch.float)
                    else:
                        rewards = torch.tensor(rewards, dtype=torch.float)
                    mean_reward = rewards.mean().item()
                    columns.append("reward")
                    columns_data.append(rewards.tolist())
                    stats[f"reward/mean{sweep_suffix}"] = mean_reward

                if self.metric_fn:
                    metric_time = time()
                    metrics = self.metric_fn(samples=str_samples, prompts=str_prompts,
                                             outputs=str_outputs, **metadata)
                    stats["time/metric"] = time() - metric_time
                    stats.update({
                        f"metrics/{k}{sweep_suffix}": torch.as_tensor(xs, dtype=torch.float).mean(-1).item()
                        for k, xs in metrics.items()
                    })
                    for metric, values in metrics.items():
                        if isinstance(values, float):
                            continue
                        columns.append(metric)
                        columns_data.append(values.tolist() if not isinstance(values, list) else values)

                if gen_sweep_value is not None:
                    columns.insert(0, gen_sweep_arg)
                    columns_data.insert(0, [gen_sweep_value] * len(str_prompts))
                table.append(list(zip(*columns_data)))

        if comm.is_main_process() and table:
            rows = sum(list(map(list, zip(*table))), [])
            title = f"Evaluation #{self.nth_evaluation}"
            for k, x in stats.items():
                if k.startswith("reward") or k.startswith("metrics"):
                    title += f" {k}: {significant(x)}"
            try:
                from rich.console import Console
                from rich.table import Table

                rich_table = Table(*columns, title=title, show_lines=True)
                for ix in range(max(min(3, len(rows)), len(gen_sweep_values))):
                    rich_table.add_row(*[str(significant(x)) for x in rows[ix]])
                Console().print(rich_table)
            except ImportError:
                logger.info(title)

        self.nth_evaluation += 1
        return stats

    # --- training loop -----------------------------------------------------------

    @contextlib.contextmanager
    def _accumulate(self):
        """no_sync for all but the last microbatch
        (reference accelerate_base_trainer.py:502-516)."""
        self.mb_count += 1
        assert self.mb_count // self.num_mb <= self.config.train.total_steps, \
            "Beyond total steps, something is wrong"
        if (self.mb_count % self.num_mb == 0
                or self.mb_count // self.num_mb >= self.config.train.total_steps):
            context = contextlib.nullcontext
            with context():
                yield
        else:
            with self.reducer.no_sync():
                yield

    def backward(self, loss: torch.Tensor):
        loss.backward()

    # --- hipGraph-captured train step --------------------------------------

    def _batch_tensors(self, batch):
        import dataclasses

        if dataclasses.is_dataclass(batch):
            return {f.name: getattr(batch, f.name) for f in dataclasses.fields(batch)}
        if isinstance(batch, dict):
            return dict(batch)
        return None

    def _graphed_loss_backward(self, microbatch):
        """Run loss forward + backward as ONE hipGraph replay.

        The train step is launch-bound on top of its kernels (~1.5 ms/step of
        host gaps at GPT-2 scale, profile r01); capturing fwd+bwd collapses
        it to a single graph launch.  Shape-keyed cache (PPO minibatches trim
        to the minibatch max response width, so a handful of shapes recur).
        Eligibility is conservative: CUDA, single process (collectives don't
        capture), no grad accumulation (capture warmup would clobber
        accumulated grads), all batch fields tensors, all loss stats device
        tensors.  Returns None to use the eager path.
        """
        if (not torch.cuda.is_available() or self.device.type != "cuda"
                or self.num_mb != 1 or self.world_size > 1
                or os.environ.get("TRLX_AMD_NO_TRAIN_GRAPH") == "1"
                or os.environ.get("TRLX_AMD_NO_GRAPHS") == "1"):
            return None
        fields = self._batch_tensors(microbatch)
        if fields is None or not all(torch.is_tensor(v) for v in fields.values()):
            return None

        if not hasattr(self, "_train_graphs"):
            self._train_graphs = {}
        key = tuple((k, tuple(v.shape), v.dtype) for k, v in sorted(fields.items()))
        entry = self._train_graphs.get(key)
        if entry is None:
            # PPO minibatches trim to the minibatch max response width, so as
            # training shifts response lengths new widths appear; with the
            # old cap of 8 a 20-step run degraded train from 85 to 108
            # ms/step once later shapes ran permanently eager.  Width count
            # is bounded by max_new_tokens; each graph holds ~100 MB of
            # static+pool at GPT-2 bench scale (288 GB HBM).
            cap = int(os.environ.get("TRLX_AMD_TRAIN_GRAPH_CAP", "32"))
            if len(self._train_graphs) >= cap:
                return None  # too many distinct shapes; stay eager
            static = {k: v.to(self.device).clone() for k, v in fields.items()}
            static_batch = type(microbatch)(**static)
            torch.cuda.synchronize()
            try:
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    for _ in range(2):
                        loss, stats = self.loss(static_batch)
                        if not all(torch.is_tensor(v) for v in stats.values()):
                            torch.cuda.current_stream().wait_stream(side)
                            self._train_graphs[key] = False
                            logger.info("train-step graph disabled: non-tensor stats %s",
                                        [k for k, v in stats.items() if not torch.is_tensor(v)])
                            return None
                        loss.backward()
                torch.cuda.current_stream().wait_stream(side)
                torch.cuda.synchronize()
                graph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(graph):
                    loss_out, stats_out = self.loss(static_batch)
                    loss_out.backward()
            except Exception as e:  # capture-unsafe loss (host sync etc)
                logger.info("train-step graph disabled for this shape: %s", e)
                torch.cuda.synchronize()
                self._train_graphs[key] = False
                self.opt.zero_grad()
                return None
            # warmup + capture polluted the grads; this is the start of an
            # optimizer window (num_mb == 1), so zeroing is safe
            self.opt.zero_grad()
            entry = (graph, static, loss_out, stats_out)
            self._train_graphs[key] = entry
            logger.info("captured train-step graph #%d for %s",
                        len([v for v in self._train_graphs.values() if v]), key[:2])
        elif entry is False:
            return None

        graph, static, loss_out, stats_out = entry
        for k, v in fields.items():
            static[k].copy_(v, non_blocking=True)
        graph.replay()
        return loss_out, stats_out

    def learn(self):
        """The main loop (reference accelerate_base_trainer.py:518-652)."""
        logger.info("Starting training")
        self.prepare_learning()
        self.iter_count = 0
        self.nth_evaluation = 0

        results = self.evaluate()
        self.tracker.log(results, step=self.iter_count)

        tbar = logging.tqdm(
            initial=self.iter_count, total=self.total_steps,
            disable=not comm.is_main_process(), position=0, leave=True,
        )
        best_reward = -float("inf")

        # train-step forward runs in eval mode (dropout off — PPO convention;
        # this was previously implicit in the per-microbatch mode toggles)
        self.model.eval()
        for _ in range(self.config.train.epochs):
            for _ in range(self.n_inner_epochs):
                train_dataloader = self.create_train_dataloader()
                for minibatch in MiniBatchIterator(train_dataloader, self.mb_size, self.num_mb):
                    forward_time = 0.0
                    backward_time = 0.0
                    stats_accum = []
                    if self.pp_size > 1:
                        # pipelined fwd+bwd over the whole minibatch (1F1B);
                        # grads accumulate inside, reduced once in finalize()
                        forward_time -= time()
                        with self.reducer.no_sync():
                            _loss, stats = self.pp_train_minibatch(list(minibatch))
                        forward_time += time()
                        stats_accum.append(stats)
                    else:
                        for microbatch in minibatch:
                            with self._accumulate():
                                graphed = self._graphed_loss_backward(microbatch)
                                if graphed is not None:
                                    forward_time -= time()
                                    loss, stats = graphed
                                    forward_time += time()
                                    stats_accum.append(stats)
                                    continue
                                forward_time -= time()
                                loss, stats = self.loss(microbatch)
                                forward_time += time()
                                backward_time -= time()
                                self.backward(loss)
                                backward_time += time()
                                stats_accum.append(stats)
                    forward_time /= self.num_mb
                    backward_time /= self.num_mb
                    stats = {k: sum(s[k] for s in stats_accum) / len(stats_accum)
                             for k in stats_accum[0]}

                    self.reducer.finalize()
                    self.opt.step()
                    self.opt.zero_grad()
                    self.scheduler.step()
                    self.iter_count += 1

                    if (self.iter_count % self.config.train.checkpoint_interval == 0
                            or self.iter_count >= self.total_steps):
                        subfolder = f"checkpoint_{self.iter_count:0{len(str(self.total_steps))}d}"
                        directory = os.path.join(self.config.train.checkpoint_dir, subfolder)
                        if self.config.train.save_optimizer:
                            self.save(directory)
                        self.save_pretrained(os.path.join(directory, "hf_model"))

                    stats["time/forward"] = forward_time
                    stats["time/backward"] = backward_time
                    for group_number, lr in enumerate(self.scheduler.get_last_lr()):
                        stats[f"learning_rate_group_{group_number}"] = lr

                    if (self.iter_count % self.config.train.eval_interval == 0
                            or self.iter_count >= self.total_steps):
                        results = self.evaluate()
                        stats.update(results)
                        if self.config.train.save_best:
                            if stats.get("reward/mean", -float("inf")) > best_reward:
                                best_reward = stats.get("reward/mean")
                                do_save = True
                            elif stats.get("metrics/reward", -float("inf")) > best_reward:
                                best_reward = stats.get("metrics/reward")
                                do_save = True
                            else:
                                do_save = False
                            do_save = comm.all_reduce_max_flag(do_save, self.device)
                            if do_save:
                                directory = os.path.join(self.config.train.checkpoint_dir, "best_checkpoint")
                                if self.config.train.save_optimizer:
                                    self.save(directory)
                                self.save_pretrained(os.path.join(directory, "hf_model"))

                    if self.iter_count % 10 == 1 or self.iter_count >= self.total_steps:
                        # formatting loss stats forces a device sync; do it
                        # every few steps, not every step
                        desc = " | ".join(f"{k}: {v:.2f}" for k, v in stats.items()
                                          if k.startswith("loss"))
                        tbar.set_description(f"[{desc}]")
                    tbar.update()
                    self.tracker.log(stats, step=self.iter_count)

                    if self.iter_count >= self.total_steps:
                        self.tracker.finish()
                        return results
                self.post_backward_callback()
            self.post_epoch_callback()
        tbar.close()
        self.tracker.finish()

    # --- abstract hooks -----------------------------------------------------------

    @abstractmethod
    def create_train_dataloader(self):
        pass

    @abstractmethod
    def get_arch(self, config: TRLConfig):
        """Build the wrapped model for this method."""
        pass

    @abstractmethod
    def loss(self, batch) -> Tuple[torch.Tensor, Dict]:
        pass

    def pp_train_minibatch(self, microbatches) -> Tuple[float, Dict]:
        """Pipelined fwd+bwd for one minibatch (list of microbatches); only
        trainers that support pipeline parallelism implement this."""
        raise NotImplementedError(
            f"{type(self).__name__} does not support pipeline_parallel_size > 1"
        )

    @abstractmethod
    def prepare_learning(self):
        pass

    def post_backward_callback(self):
        pass

    def post_epoch_callback(self):
        pass
