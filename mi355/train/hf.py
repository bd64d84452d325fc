"""HF-Trainer-compatible inner loop (reference cell-34 NLP /
Scaling_model_training.ipynb:cell-47 build a transformers.Trainer inside
trainer_init_per_worker). Same constructor surface and TrainingArguments
fields the reference uses; internals are MI355X-native: FlatReplica DDP
(RCCL/xGMI), fused-AdamW flat optimizer, bf16 when requested."""
from __future__ import annotations

import os
from dataclasses import dataclass, field

import torch

from ..parallel import (FlatAdamW, FlatReplica, get_rank, get_world_size,
                        is_distributed)
from . import session


@dataclass
class TrainingArguments:
    output_dir: str = "output"
    learning_rate: float = 5e-5
    num_train_epochs: float = 3.0
    weight_decay: float = 0.0
    per_device_train_batch_size: int = 8
    per_device_eval_batch_size: int = 8
    evaluation_strategy: str = "no"      # "no" | "epoch" | "steps"
    save_strategy: str = "epoch"
    logging_strategy: str = "epoch"
    eval_steps: int | None = None
    save_steps: int | None = None
    logging_steps: int = 500
    max_grad_norm: float = 1.0
    adam_beta1: float = 0.9
    adam_beta2: float = 0.999
    adam_epsilon: float = 1e-8
    lr_scheduler_type: str = "linear"
    warmup_steps: int = 0
    warmup_ratio: float = 0.0
    bf16: bool = False
    fp16: bool = False
    no_cuda: bool = False
    seed: int = 42
    disable_tqdm: bool = True
    remove_unused_columns: bool = True
    eval_accumulation_steps: int | None = None
    gradient_accumulation_steps: int = 1
    save_total_limit: int | None = None
    log_level: str = "passive"
    log_on_each_node: bool = True
    push_to_hub: bool = False
    max_steps: int = -1

    def __post_init__(self):
        os.makedirs(self.output_dir, exist_ok=True)


class Trainer:
    """Constructor-compatible subset of transformers.Trainer covering every
    field the reference uses."""

    def __init__(self, model=None, args: TrainingArguments | None = None,
                 train_dataset=None, eval_dataset=None, tokenizer=None,
                 data_collator=None, compute_metrics=None,
                 optimizers=(None, None), preprocess_logits_for_metrics=None):
        self.model = model
        self.args = args or TrainingArguments()
        self.train_dataset = train_dataset
        self.eval_dataset = eval_dataset
        self.tokenizer = tokenizer
        self.compute_metrics = compute_metrics
        self._opt_override, self._sched_override = optimizers
        self.state = type("TrainerState", (), {"epoch": 0.0, "global_step": 0})()

    # ------------------------------------------------------------------ #
    def _device_dtype(self):
        use_cuda = torch.cuda.is_available() and not self.args.no_cuda
        device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", "0")) %
                              max(torch.cuda.device_count(), 1)) if use_cuda \
            else torch.device("cpu")
        if self.args.bf16:
            dtype = torch.bfloat16
        elif self.args.fp16:
            dtype = torch.float16
        else:
            dtype = torch.bfloat16 if use_cuda else torch.float32
        return device, dtype

    def _iter_batches(self, dataset, batch_size, device, shuffle_seed=None):
        from ..data import Dataset

        if dataset is None:
            return
        if isinstance(dataset, Dataset):
            if shuffle_seed is not None:
                dataset = dataset.random_shuffle(seed=shuffle_seed)
            yield from dataset.iter_torch_batches(batch_size=batch_size, device=device)
            return
        # list-of-dicts / torch datasets
        items = list(dataset)
        for i in range(0, len(items), batch_size):
            chunk = items[i:i + batch_size]
            batch = {k: torch.as_tensor([c[k] for c in chunk]).to(device)
                     for k in chunk[0]}
            yield batch

    def _model_inputs(self, batch):
        import inspect

        if not self.args.remove_unused_columns:
            return batch
        sig = inspect.signature(self.model.forward)
        return {k: v for k, v in batch.items() if k in sig.parameters}

    # ------------------------------------------------------------------ #
    def train(self, resume_from_checkpoint=None):
        args = self.args
        torch.manual_seed(args.seed)
        device, dtype = self._device_dtype()
        self.model = self.model.to(device=device, dtype=dtype)
        self.model.train()

        replica = FlatReplica(self.model)
        if self._opt_override is not None:
            optimizer = self._opt_override
            scheduler = self._sched_override
            fused = False
        else:
            optimizer = FlatAdamW(replica, lr=args.learning_rate,
                                  betas=(args.adam_beta1, args.adam_beta2),
                                  eps=args.adam_epsilon,
                                  weight_decay=args.weight_decay,
                                  max_grad_norm=args.max_grad_norm)
            scheduler = None
            fused = True
        self._replica = replica

        n_epochs = int(args.num_train_epochs)
        steps_per_epoch = None
        total_steps = args.max_steps if args.max_steps > 0 else None
        global_step = 0
        history = []
        stopped = False
        for epoch in range(n_epochs):
            if stopped:
                break
            epoch_loss, nb = 0.0, 0
            for batch in self._iter_batches(self.train_dataset,
                                            args.per_device_train_batch_size,
                                            device, shuffle_seed=args.seed + epoch):
                inputs = self._model_inputs(batch)
                # zero BEFORE forward: zero_grads also advances the fused
                # dropout seed, and fwd+bwd of one step must see the same masks
                if fused:
                    optimizer.zero_grad()
                else:
                    replica.zero_grads()
                out = self.model(**inputs)
                loss = out.loss if hasattr(out, "loss") else out["loss"]
                loss.backward()
                replica.finish_grad_sync()
                if fused:
                    lr = self._lr_at(global_step, total_steps, steps_per_epoch,
                                     n_epochs)
                    optimizer.step(lr=lr)
                else:
                    if args.max_grad_norm:
                        torch.nn.utils.clip_grad_norm_(self.model.parameters(),
                                                       args.max_grad_norm)
                    optimizer.step()
                    if scheduler is not None:
                        scheduler.step()
                    optimizer.zero_grad()
                epoch_loss += float(loss.detach().float().cpu())
                nb += 1
                global_step += 1
                if total_steps and global_step >= total_steps:
                    break
            if steps_per_epoch is None:
                steps_per_epoch = max(nb, 1)
            self.state.epoch = epoch + 1.0
            self.state.global_step = global_step

            metrics = {"epoch": epoch + 1, "step": global_step,
                       "loss": epoch_loss / max(nb, 1),
                       "learning_rate": self._lr_at(global_step, total_steps,
                                                    steps_per_epoch, n_epochs)}
            if args.evaluation_strategy == "epoch" and self.eval_dataset is not None:
                metrics.update(self.evaluate())
                self.model.train()
            ckpt_dir = None
            if args.save_strategy == "epoch" and get_rank() == 0:
                ckpt_dir = os.path.join(args.output_dir, f"checkpoint-{global_step}")
                self.save_model(ckpt_dir)
            if is_distributed():
                import torch.distributed as dist

                dist.barrier()  # M5: rank0 writes, others wait
            history.append(metrics)
            if not session.report(metrics, checkpoint=ckpt_dir):
                stopped = True  # ASHA early stop
        return type("TrainOutput", (), {"global_step": global_step,
                                        "metrics": history[-1] if history else {},
                                        "history": history})()

    def _lr_at(self, step, total_steps, steps_per_epoch, n_epochs):
        args = self.args
        total = total_steps or ((steps_per_epoch or 100) * n_epochs)
        warmup = args.warmup_steps or int(args.warmup_ratio * total)
        if warmup and step < warmup:
            return args.learning_rate * step / max(warmup, 1)
        if args.lr_scheduler_type == "constant":
            return args.learning_rate
        frac = min(max(total - step, 0) / max(total - warmup, 1), 1.0)
        return args.learning_rate * frac

    @torch.no_grad()
    def evaluate(self, eval_dataset=None) -> dict:
        ds = eval_dataset or self.eval_dataset
        device, _ = self._device_dtype()
        self.model.eval()
        total, nb = 0.0, 0
        for batch in self._iter_batches(ds, self.args.per_device_eval_batch_size,
                                        device):
            out = self.model(**self._model_inputs(batch))
            loss = out.loss if hasattr(out, "loss") else out["loss"]
            total += float(loss.detach().float().cpu())
            nb += 1
        mean = total / max(nb, 1)
        if is_distributed():  # M4: aggregate eval metric across workers
            import torch.distributed as dist

            # tensor must live on the backend's device (RCCL = cuda)
            t = torch.tensor([mean * nb, float(nb)], device=device)
            dist.all_reduce(t)
            mean = float(t[0] / max(float(t[1]), 1.0))
        return {"eval_loss": mean}

    def save_model(self, output_dir: str):
        os.makedirs(output_dir, exist_ok=True)
        if hasattr(self.model, "save_pretrained"):
            self.model.save_pretrained(output_dir)
        else:
            torch.save(self.model.state_dict(),
                       os.path.join(output_dir, "pytorch_model.bin"))
        if self.tokenizer is not None and hasattr(self.tokenizer, "save_pretrained"):
            self.tokenizer.save_pretrained(output_dir)
