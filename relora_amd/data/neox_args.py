"""Lean NeoXArgs: the Megatron-config surface the ReLoRA data path consumes.

The reference carries a ~2800-LoC frozen-dataclass stack
(peft_pretraining/megatron_dataset/arguments.py, neox_args.py) of which the
training path reads only the data/batch fields.  This
class accepts the same YAML dicts (unknown keys are stored but unused),
applies the same defaults for the consumed fields, and reproduces
``calculate_batch_parameters``'s solver semantics (arguments.py:753-791).
"""

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from relora_amd.utils.logging import logger


def _none_if_blank(v):
    return None if v in ("", None) else v


@dataclass
class NeoXArgs:
    # data (defaults mirror reference neox_args.py)
    data_path: Optional[str] = None
    train_data_paths: Optional[List[str]] = None
    valid_data_paths: Optional[List[str]] = None
    test_data_paths: Optional[List[str]] = None
    label_data_paths: Optional[List[str]] = None
    train_data_weights: Optional[List[float]] = None
    valid_data_weights: Optional[List[float]] = None
    test_data_weights: Optional[List[float]] = None
    weight_by_num_documents: bool = False
    weighted_sampler_alpha: float = 0.3
    use_shared_fs: bool = True
    data_impl: str = "infer"
    mmap_warmup: bool = False
    split: str = "969, 30, 1"
    seq_length: int = 2048
    seed: int = 1234
    num_workers: int = 2

    # schedule
    train_iters: Optional[int] = None
    eval_iters: int = 100
    eval_interval: int = 1000
    iteration: Optional[int] = None

    # batch algebra
    global_num_gpus: Optional[int] = None
    train_batch_size: Optional[int] = None
    train_micro_batch_size_per_gpu: Optional[int] = None
    gradient_accumulation_steps: Optional[int] = None

    # parallel-config compat (config-only in the reference too)
    pipe_parallel_size: int = 0
    model_parallel_size: int = 1

    # tokenizer
    tokenizer_type: str = "HFTokenizer"
    vocab_file: Optional[str] = None

    # runtime flags filled by build_train_valid_test_dataloaders
    do_train: int = 0
    do_valid: int = 0
    do_test: int = 0

    # everything else from the YAML, kept for introspection
    extra_args: Dict[str, Any] = field(default_factory=dict)

    # type table for the consumed fields — the validation the reference's
    # dataclass stack applies per-field (arguments.py:109-1240), scoped to
    # the surface this trainer reads
    _FIELD_TYPES = {
        "data_path": str, "train_data_paths": list, "valid_data_paths": list,
        "test_data_paths": list, "label_data_paths": list,
        "train_data_weights": list, "valid_data_weights": list,
        "test_data_weights": list, "weight_by_num_documents": bool,
        "weighted_sampler_alpha": float, "use_shared_fs": bool,
        "data_impl": str, "mmap_warmup": bool, "split": str,
        "seq_length": int, "seed": int, "num_workers": int,
        "train_iters": int, "eval_iters": int, "eval_interval": int,
        "iteration": int, "global_num_gpus": int, "train_batch_size": int,
        "train_micro_batch_size_per_gpu": int,
        "gradient_accumulation_steps": int, "pipe_parallel_size": int,
        "model_parallel_size": int, "tokenizer_type": str, "vocab_file": str,
    }

    @classmethod
    def _check_type(cls, k, v):
        want = cls._FIELD_TYPES.get(k)
        if want is None or v is None:
            return v
        if want is float and isinstance(v, (int, float)) and not isinstance(v, bool):
            return float(v)
        if want is int and isinstance(v, int) and not isinstance(v, bool):
            return v
        if want is bool and isinstance(v, bool):
            return v
        if want in (str, list) and isinstance(v, want):
            return v
        raise ValueError(
            f"NeoXArgs: field {k!r} expects {want.__name__}, got "
            f"{type(v).__name__} ({v!r})")

    @classmethod
    def from_dict(cls, d: Dict[str, Any], strict: Optional[bool] = None):
        """Build from a YAML dict.  Consumed fields are type-checked; unknown
        keys are kept in `extra_args` with a warning (they are reference
        NeoXArgs fields this trainer does not consume), or rejected when
        `strict=True` / RELORA_AMD_NEOX_STRICT=1 — the fail-on-unknown-key
        mode of the reference's validation stack."""
        import os
        if strict is None:
            strict = os.environ.get("RELORA_AMD_NEOX_STRICT", "0") == "1"
        known = {f for f in cls.__dataclass_fields__ if f != "extra_args"}
        kwargs, extra = {}, {}
        for k, v in d.items():
            if k in known:
                kwargs[k] = cls._check_type(k, _none_if_blank(v))
            else:
                extra[k] = v
        if extra:
            if strict:
                raise ValueError(
                    f"NeoXArgs(strict): unknown config keys {sorted(extra)}")
            logger.warning(
                f"NeoXArgs: {len(extra)} config keys not consumed by this "
                f"trainer (kept in extra_args): {sorted(extra)[:12]}"
                f"{' ...' if len(extra) > 12 else ''}")
        args = cls(extra_args=extra, **kwargs)
        args.calculate_derived()
        return args

    # -- derived values ----------------------------------------------------

    @property
    def is_pipe_parallel(self):
        return self.pipe_parallel_size > 1

    @property
    def batch_size(self):
        """Micro-batch per GPU (the name the data loader uses)."""
        return self.train_micro_batch_size_per_gpu

    @staticmethod
    def calculate_batch_parameters(dp_world_size, train_batch=None,
                                   micro_batch=None, grad_acc=None):
        """Solve the missing member(s) of train = micro × grad_acc × world."""
        if train_batch is not None and micro_batch is not None and grad_acc is not None:
            return int(train_batch), int(micro_batch), int(grad_acc)
        if train_batch is not None and micro_batch is not None:
            grad_acc = train_batch // micro_batch // dp_world_size
        elif train_batch is not None and grad_acc is not None:
            micro_batch = train_batch // dp_world_size // grad_acc
        elif micro_batch is not None and grad_acc is not None:
            train_batch = micro_batch * grad_acc * dp_world_size
        elif train_batch is not None:
            grad_acc = 1
            micro_batch = train_batch // dp_world_size
        elif micro_batch is not None:
            train_batch = micro_batch * dp_world_size
            grad_acc = 1
        else:
            raise ValueError(
                "either train_batch_size or train_micro_batch_size_per_gpu is required")
        return int(train_batch), int(micro_batch), int(grad_acc)

    def calculate_derived(self):
        if self.global_num_gpus is None:
            raise RuntimeError(
                "global_num_gpus must be provided (the trainer injects world_size)")
        mp = self.model_parallel_size or 1
        pp = self.pipe_parallel_size or 1
        dp_world_size = self.global_num_gpus // max(pp, 1) // mp
        if dp_world_size * max(pp, 1) * mp != self.global_num_gpus:
            raise ValueError(
                f"global_num_gpus={self.global_num_gpus} not divisible by "
                f"pipe({pp}) x model({mp}) parallel sizes")

        train, micro, acc = self.calculate_batch_parameters(
            dp_world_size, self.train_batch_size,
            self.train_micro_batch_size_per_gpu, self.gradient_accumulation_steps)
        self.train_batch_size, self.train_micro_batch_size_per_gpu = train, micro
        self.gradient_accumulation_steps = acc

        if not all(x > 0 for x in (train, micro, acc)):
            raise ValueError("batch sizes must be positive")
        if train != micro * acc * dp_world_size:
            raise ValueError(
                f"train_batch_size {train} != micro {micro} x grad_acc {acc} "
                f"x dp_world {dp_world_size}")

        if self.train_data_paths and self.train_data_weights is None:
            self.train_data_weights = [1.0] * len(self.train_data_paths)
        if self.valid_data_paths and self.valid_data_weights is None:
            self.valid_data_weights = [1.0] * len(self.valid_data_paths)
        if self.test_data_paths and self.test_data_weights is None:
            self.test_data_weights = [1.0] * len(self.test_data_paths)

        if self.data_path is None and not self.train_data_paths:
            raise ValueError("either data_path or train_data_paths is required")
        if self.extra_args:
            logger.debug(f"NeoXArgs: {len(self.extra_args)} unused config keys kept "
                         f"in extra_args")
        return self
