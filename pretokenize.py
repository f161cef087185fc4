"""Download and pre-tokenize a huggingface dataset into the on-disk layout
the trainer's `--dataset_path` expects:
`{save_dir}/{dataset}_{tokenizer}_{seqlen}/` + `args.json`
(parity with reference `pretokenize.py:22-88`; requires network access for
remote datasets — also accepts local dataset paths)."""

import argparse
import json
import multiprocessing
import os
import time

from relora_amd.utils.logging import logger


def parse_args(args=None):
    parser = argparse.ArgumentParser()
    parser.add_argument("--tokenizer", type=str, required=True)
    parser.add_argument("--dataset", type=str, required=True)
    parser.add_argument("--dataset_config", type=str, default=None)
    parser.add_argument("--text_field", type=str, default="text")
    parser.add_argument("--sequence_length", type=int, default=2048)
    parser.add_argument("--num_cpu", type=int, default=multiprocessing.cpu_count())
    parser.add_argument("--save_dir", type=str, required=True)
    parser.add_argument("--take", type=int, default=None)
    return parser.parse_args(args)


def main(args):
    from datasets import Dataset, DatasetDict, load_dataset, load_from_disk
    from transformers import AutoTokenizer

    from relora_amd.data.dataloader import tokenize_and_chunk

    _tokenizer_name_for_save = args.tokenizer.replace("/", "_")
    save_path = os.path.join(
        args.save_dir, f"{args.dataset.replace('/', '_')}_{_tokenizer_name_for_save}_{args.sequence_length}"
    )
    if args.dataset_config is not None:
        save_path = os.path.join(
            args.save_dir,
            f"{args.dataset.replace('/', '_')}_{args.dataset_config}_{_tokenizer_name_for_save}_{args.sequence_length}",
        )
    if os.path.exists(save_path):
        raise ValueError(f"Path {save_path} already exists")

    tokenizer = AutoTokenizer.from_pretrained(args.tokenizer)
    if os.path.isdir(args.dataset):
        dataset = load_from_disk(args.dataset)
    else:
        dataset = load_dataset(args.dataset, args.dataset_config, streaming=args.take is not None)

    if args.take is not None:
        logger.info(f"Taking {args.take} examples from the dataset")

        def take(ds, n):
            return Dataset.from_generator(lambda: (yield from ds.take(n)))

        dataset = DatasetDict({k: take(v, args.take) for k, v in dataset.items()})

    _time = time.time()
    dataset = tokenize_and_chunk(
        tokenizer=tokenizer,
        dataset=dataset,
        text_field=args.text_field,
        sequence_length=args.sequence_length,
        num_cpu=args.num_cpu,
    )
    logger.info(f"Tokenization and chunking took {(time.time() - _time) / 3600:.2f} hours")

    dataset.save_to_disk(save_path)
    logger.info(f"Saved the dataset to {save_path}")

    with open(os.path.join(save_path, "args.json"), "w") as f:
        json.dump(vars(args), f, indent=4)


if __name__ == "__main__":
    main(parse_args())
