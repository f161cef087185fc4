"""GLUE finetuning evaluation for pretrained (Re)LoRA checkpoints.

Parity with the reference `run_glue.py` (:209-623): HfArgumentParser CLI
(ModelArguments / DataTrainingArguments / HF TrainingArguments), the nine
GLUE tasks or user csv/json files, the local `LlamaForSequenceClassification`
loaded fresh or from a pretraining checkpoint (reference :379-390).

Offline adaptations for this environment (no network, no `evaluate`):
* datasets load from `--dataset_path` (a saved-to-disk dataset) or local
  csv/json files; the hub path is attempted last;
* metrics (accuracy, F1, Matthews corr, Pearson/Spearman) are computed with
  scikit-learn / scipy instead of the `evaluate` hub scripts.

Usage:
    python run_glue.py --task_name cola --dataset_path /data/glue_cola \
        --model_name_or_path checkpoints/model_5000 --tokenizer_name t5-base \
        --output_dir out --do_train --do_eval
"""

import logging
import os
import random
import sys
from dataclasses import dataclass, field
from typing import Optional

import numpy as np

import torch  # noqa: F401  (ensures backend init before transformers)
from transformers import (
    AutoTokenizer,
    DataCollatorWithPadding,
    EvalPrediction,
    HfArgumentParser,
    Trainer,
    TrainingArguments,
    default_data_collator,
    set_seed,
)

from relora_amd.models.config import LlamaConfig, load_model_config
from relora_amd.models.llama import LlamaForSequenceClassification

logger = logging.getLogger(__name__)

task_to_keys = {
    "cola": ("sentence", None),
    "mnli": ("premise", "hypothesis"),
    "mrpc": ("sentence1", "sentence2"),
    "qnli": ("question", "sentence"),
    "qqp": ("question1", "question2"),
    "rte": ("sentence1", "sentence2"),
    "sst2": ("sentence", None),
    "stsb": ("sentence1", "sentence2"),
    "wnli": ("sentence1", "sentence2"),
}


@dataclass
class DataTrainingArguments:
    task_name: Optional[str] = field(default=None, metadata={"help": "GLUE task name"})
    dataset_path: Optional[str] = field(
        default=None, metadata={"help": "datasets.load_from_disk directory (offline)"})
    max_seq_length: int = field(default=128)
    pad_to_max_length: bool = field(default=True)
    max_train_samples: Optional[int] = field(default=None)
    max_eval_samples: Optional[int] = field(default=None)
    max_predict_samples: Optional[int] = field(default=None)
    train_file: Optional[str] = field(default=None)
    validation_file: Optional[str] = field(default=None)
    test_file: Optional[str] = field(default=None)

    def __post_init__(self):
        if self.task_name is not None:
            self.task_name = self.task_name.lower()
            if self.task_name not in task_to_keys:
                raise ValueError(
                    f"Unknown task {self.task_name}; pick one of {list(task_to_keys)}")
        elif self.dataset_path is None and (self.train_file is None or self.validation_file is None):
            raise ValueError("need a GLUE task, a dataset_path, or train/validation files")
        for f_ in (self.train_file, self.validation_file):
            if f_ is not None:
                ext = f_.split(".")[-1]
                assert ext in ("csv", "json"), "train/validation files must be csv or json"


@dataclass
class ModelArguments:
    model_name_or_path: Optional[str] = field(
        default=None, metadata={"help": "pretraining checkpoint dir (config.json + weights)"})
    model_config: Optional[str] = field(
        default=None, metadata={"help": "architecture json (fresh model, e.g. configs/llama_250m.json)"})
    tokenizer_name: Optional[str] = field(default=None)
    cache_dir: Optional[str] = field(default=None)
    use_fast_tokenizer: bool = field(default=True)
    ignore_mismatched_sizes: bool = field(default=False)


def glue_metrics(task_name, preds, labels, is_regression):
    """accuracy / F1 / Matthews / Pearson+Spearman without the evaluate hub."""
    from scipy.stats import pearsonr, spearmanr
    from sklearn.metrics import accuracy_score, f1_score, matthews_corrcoef

    if is_regression:
        return {
            "pearson": float(pearsonr(preds, labels)[0]),
            "spearmanr": float(spearmanr(preds, labels)[0]),
        }
    out = {"accuracy": float(accuracy_score(labels, preds))}
    if task_name in ("mrpc", "qqp"):
        out["f1"] = float(f1_score(labels, preds))
        out["combined_score"] = (out["accuracy"] + out["f1"]) / 2
    if task_name == "cola":
        out["matthews_correlation"] = float(matthews_corrcoef(labels, preds))
    return out


def load_raw_datasets(data_args, model_args):
    import datasets

    if data_args.dataset_path is not None:
        return datasets.load_from_disk(data_args.dataset_path)
    if data_args.train_file is not None:
        data_files = {"train": data_args.train_file,
                      "validation": data_args.validation_file}
        if data_args.test_file is not None:
            data_files["test"] = data_args.test_file
        fmt = "csv" if data_args.train_file.endswith(".csv") else "json"
        return datasets.load_dataset(fmt, data_files=data_files,
                                     cache_dir=model_args.cache_dir)
    # hub path (requires network; last resort)
    return datasets.load_dataset("glue", data_args.task_name,
                                 cache_dir=model_args.cache_dir)


def build_model(model_args, num_labels, task_name):
    if model_args.model_name_or_path:
        cfg_path = os.path.join(model_args.model_name_or_path, "config.json")
        config = load_model_config(cfg_path)
        config.num_labels = num_labels
        config.finetuning_task = task_name
        model = LlamaForSequenceClassification(config)
        weights = os.path.join(model_args.model_name_or_path, "pytorch_model.bin")
        if os.path.exists(weights):
            state = torch.load(weights, map_location="cpu", weights_only=True)
            state = {k: v for k, v in state.items() if not k.startswith("lm_head")}
            missing, unexpected = model.load_state_dict(state, strict=False)
            logger.info(f"loaded {weights}: {len(missing)} missing, "
                        f"{len(unexpected)} unexpected keys")
    else:
        config = load_model_config(model_args.model_config)
        config.num_labels = num_labels
        config.finetuning_task = task_name
        model = LlamaForSequenceClassification(config)
    return model, config


def main():
    parser = HfArgumentParser((ModelArguments, DataTrainingArguments, TrainingArguments))
    if len(sys.argv) == 2 and sys.argv[1].endswith(".json"):
        model_args, data_args, training_args = parser.parse_json_file(
            json_file=os.path.abspath(sys.argv[1]))
    else:
        model_args, data_args, training_args = parser.parse_args_into_dataclasses()

    logging.basicConfig(
        format="%(asctime)s - %(levelname)s - %(name)s - %(message)s",
        datefmt="%m/%d/%Y %H:%M:%S",
        handlers=[logging.StreamHandler(sys.stdout)],
        level=logging.INFO,
    )
    set_seed(training_args.seed)

    raw_datasets = load_raw_datasets(data_args, model_args)

    # label space
    if data_args.task_name is not None:
        is_regression = data_args.task_name == "stsb"
        if not is_regression:
            label_list = raw_datasets["train"].features["label"].names
            num_labels = len(label_list)
        else:
            label_list = None
            num_labels = 1
    else:
        is_regression = raw_datasets["train"].features["label"].dtype in ("float32", "float64")
        if is_regression:
            label_list, num_labels = None, 1
        else:
            label_list = sorted(raw_datasets["train"].unique("label"))
            num_labels = len(label_list)

    model, config = build_model(model_args, num_labels, data_args.task_name)

    tok_src = model_args.tokenizer_name or model_args.model_name_or_path
    tokenizer = AutoTokenizer.from_pretrained(
        tok_src, cache_dir=model_args.cache_dir, use_fast=model_args.use_fast_tokenizer,
        model_max_length=data_args.max_seq_length)
    if tokenizer.pad_token is None:
        tokenizer.pad_token = tokenizer.eos_token
    config.pad_token_id = tokenizer.pad_token_id

    if data_args.task_name is not None:
        sentence1_key, sentence2_key = task_to_keys[data_args.task_name]
    else:
        cols = [c for c in raw_datasets["train"].column_names if c != "label"]
        sentence1_key = cols[0]
        sentence2_key = cols[1] if len(cols) > 1 else None

    padding = "max_length" if data_args.pad_to_max_length else False
    label_to_id = None
    if label_list is not None and not is_regression and data_args.task_name is None:
        label_to_id = {v: i for i, v in enumerate(label_list)}

    max_seq_length = min(data_args.max_seq_length, config.max_position_embeddings)

    def preprocess(examples):
        args_ = ((examples[sentence1_key],) if sentence2_key is None
                 else (examples[sentence1_key], examples[sentence2_key]))
        result = tokenizer(*args_, padding=padding, max_length=max_seq_length,
                           truncation=True)
        if label_to_id is not None and "label" in examples:
            result["label"] = [(label_to_id[x] if x != -1 else -1) for x in examples["label"]]
        return result

    with training_args.main_process_first(desc="dataset map pre-processing"):
        raw_datasets = raw_datasets.map(preprocess, batched=True,
                                        desc="Running tokenizer on dataset")

    train_dataset = eval_dataset = predict_dataset = None
    if training_args.do_train:
        train_dataset = raw_datasets["train"]
        if data_args.max_train_samples is not None:
            train_dataset = train_dataset.select(
                range(min(len(train_dataset), data_args.max_train_samples)))
    if training_args.do_eval:
        key = "validation_matched" if data_args.task_name == "mnli" else "validation"
        eval_dataset = raw_datasets[key]
        if data_args.max_eval_samples is not None:
            eval_dataset = eval_dataset.select(
                range(min(len(eval_dataset), data_args.max_eval_samples)))
    if training_args.do_predict and "test" in raw_datasets:
        key = "test_matched" if data_args.task_name == "mnli" else "test"
        predict_dataset = raw_datasets[key]
        if data_args.max_predict_samples is not None:
            predict_dataset = predict_dataset.select(
                range(min(len(predict_dataset), data_args.max_predict_samples)))

    if train_dataset is not None:
        for index in random.sample(range(len(train_dataset)), min(3, len(train_dataset))):
            logger.info(f"Sample {index} of the training set: {train_dataset[index]}.")

    def compute_metrics(p: EvalPrediction):
        preds = p.predictions[0] if isinstance(p.predictions, tuple) else p.predictions
        preds = np.squeeze(preds) if is_regression else np.argmax(preds, axis=1)
        return glue_metrics(data_args.task_name, preds, p.label_ids, is_regression)

    if data_args.pad_to_max_length:
        data_collator = default_data_collator
    elif training_args.fp16:
        data_collator = DataCollatorWithPadding(tokenizer, pad_to_multiple_of=8)
    else:
        data_collator = None

    trainer = Trainer(
        model=model,
        args=training_args,
        train_dataset=train_dataset,
        eval_dataset=eval_dataset,
        compute_metrics=compute_metrics,
        processing_class=tokenizer,
        data_collator=data_collator,
    )

    if training_args.do_train:
        result = trainer.train(resume_from_checkpoint=training_args.resume_from_checkpoint)
        trainer.save_model()
        trainer.log_metrics("train", result.metrics)
        trainer.save_metrics("train", result.metrics)
        trainer.save_state()

    if training_args.do_eval:
        logger.info("*** Evaluate ***")
        metrics = trainer.evaluate(eval_dataset=eval_dataset)
        trainer.log_metrics("eval", metrics)
        trainer.save_metrics("eval", metrics)

    if training_args.do_predict and predict_dataset is not None:
        logger.info("*** Predict ***")
        predict_dataset = predict_dataset.remove_columns("label")
        preds = trainer.predict(predict_dataset, metric_key_prefix="predict").predictions
        preds = np.squeeze(preds) if is_regression else np.argmax(preds, axis=1)
        out_file = os.path.join(training_args.output_dir,
                                f"predict_results_{data_args.task_name}.txt")
        if trainer.is_world_process_zero():
            with open(out_file, "w") as writer:
                writer.write("index\tprediction\n")
                for index, item in enumerate(preds):
                    if is_regression:
                        writer.write(f"{index}\t{item:3.3f}\n")
                    else:
                        item = label_list[item] if label_list else item
                        writer.write(f"{index}\t{item}\n")


if __name__ == "__main__":
    main()
