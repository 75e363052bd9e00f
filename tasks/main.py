#!/usr/bin/env python3
"""Downstream-task finetune/eval driver (reference tasks/main.py).

Dispatches on --task:
  MNLI / QQP      — GLUE classification finetune (TSV data)
  RACE            — multiple-choice finetune (JSON-lines data)
  WIKITEXT103     — zero-shot GPT perplexity (text file)
  LAMBADA         — zero-shot GPT cloze accuracy (JSON-lines)

Text is tokenized with the built-in byte tokenizer so the harness runs
without vocab files; --tokenizer-type is honored for GPT eval when ids
are pre-tokenized.

  torchrun --nproc-per-node 1 --master-addr 127.0.0.1 tasks/main.py \
      --task MNLI --train-data train.tsv --valid-data dev.tsv \
      --num-layers 4 --hidden-size 256 --num-attention-heads 8 \
      --seq-length 128 --micro-batch-size 8 --global-batch-size 8 \
      --epochs 3 --lr 1e-5
"""

import os
import sys

sys.path.append(os.path.abspath(
    os.path.join(os.path.dirname(__file__), os.path.pardir)))

from megatronapp_amd.training.global_vars import get_args
from megatronapp_amd.training.initialize import initialize_megatron


def get_tasks_args(parser):
    g = parser.add_argument_group("tasks")
    g.add_argument("--task", type=str, required=True)
    g.add_argument("--epochs", type=int, default=None)
    g.add_argument("--train-data", nargs="+", default=None)
    g.add_argument("--valid-data", nargs="*", default=None)
    g.add_argument("--overlapping-eval", type=int, default=32)
    g.add_argument("--pretrained-checkpoint", type=str, default=None)
    g.add_argument("--qa-data", type=str, default=None)
    g.add_argument("--evidence-data", type=str, default=None)
    g.add_argument("--biencoder-projection-dim", type=int, default=0)
    g.add_argument("--prompt-type", choices=["knowledge", "response"],
                   default="knowledge")
    g.add_argument("--sample-input-file", type=str, default=None)
    g.add_argument("--sample-output-file", type=str, default=None)
    g.add_argument("--guess-file", type=str, default=None)
    g.add_argument("--answer-file", type=str, default=None)
    g.add_argument("--out-seq-length", type=int, default=64)
    return parser


def main():
    initialize_megatron(
        extra_args_provider=get_tasks_args,
        args_defaults={"tokenizer_type": "NullTokenizer",
                       "vocab_size": 259})
    args = get_args()
    from tasks.data_utils import ByteTokenizer
    tokenizer = ByteTokenizer()

    task = args.task.upper()
    if task in ("MNLI", "QQP"):
        from tasks.finetune_utils import finetune
        from tasks.glue.data import MNLIDataset, QQPDataset
        cls = MNLIDataset if task == "MNLI" else QQPDataset
        train = cls("training", args.train_data, tokenizer,
                    args.seq_length)
        valid = cls("validation", args.valid_data, tokenizer,
                    args.seq_length)
        finetune(train, valid, cls.num_classes, name=task)
    elif task == "RACE":
        from tasks.finetune_utils import finetune
        from tasks.race.data import RaceDataset
        train = RaceDataset("training", args.train_data, tokenizer,
                            args.seq_length)
        valid = RaceDataset("validation", args.valid_data, tokenizer,
                            args.seq_length)
        finetune(train, valid, RaceDataset.num_classes, name=task)
    elif task in ("WIKITEXT103", "LAMBADA"):
        from tasks.zeroshot_gpt.evaluate import main as zeroshot_main
        zeroshot_main(tokenizer)
    elif task in ("ORQA", "NQ"):
        from tasks.orqa.evaluate import main as orqa_main
        orqa_main(tokenizer)
    elif task in ("MSDP-PROMPT",):
        import json
        from tasks.msdp.prompt import run_prompting
        from tasks.zeroshot_gpt.evaluate import build_gpt
        model, device = build_gpt(args)
        with open(args.sample_input_file, encoding="utf-8") as f:
            samples = [json.loads(line) for line in f if line.strip()]
        run_prompting(model, tokenizer, samples, args.prompt_type,
                      args.out_seq_length, device,
                      args.sample_output_file)
        print(f"wrote {len(samples)} generations to "
              f"{args.sample_output_file}", flush=True)
    elif task in ("MSDP-EVAL-F1",):
        from tasks.msdp.evaluate import evaluate_f1
        evaluate_f1(args.guess_file, args.answer_file)
    else:
        raise NotImplementedError(f"task {args.task} is not implemented "
                                  "(available: MNLI QQP RACE WIKITEXT103 "
                                  "LAMBADA ORQA MSDP-PROMPT MSDP-EVAL-F1)")


if __name__ == "__main__":
    main()
