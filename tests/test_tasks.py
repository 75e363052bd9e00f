"""Downstream-task harness: GLUE/RACE finetune + zero-shot GPT eval
(reference tasks/)."""
import json
import os
import subprocess
import sys

import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _write_mnli(path, n=12):
    rows = ["idx\t1\t2\t3\t4\t5\t6\t7\tsentence1\tsentence2\tlabel"]
    labels = ["contradiction", "entailment", "neutral"]
    for i in range(n):
        lab = labels[i % 3]
        rows.append(f"{i}\tx\tx\tx\tx\tx\tx\tx\tcats sit {lab}\t"
                    f"dogs run {i}\t{lab}")
    path.write_text("\n".join(rows) + "\n")


def _write_qqp(path, n=12):
    rows = ["id\tqid1\tqid2\tquestion1\tquestion2\tis_duplicate"]
    for i in range(n):
        rows.append(f"{i}\ta\tb\thow to {i}?\twhat is {i}?\t{i % 2}")
    path.write_text("\n".join(rows) + "\n")


def _write_race(path, n=6):
    lines = []
    for i in range(n):
        lines.append(json.dumps({
            "article": f"The answer to question {i} is option "
                       f"{chr(ord('A') + i % 4)}.",
            "questions": [f"Which option _ is right for {i}?"],
            "options": [["A", "B", "C", "D"]],
            "answers": [chr(ord("A") + i % 4)],
        }))
    path.write_text("\n".join(lines) + "\n")


def test_byte_tokenizer_and_build_sample():
    from tasks.data_utils import ByteTokenizer, build_sample
    tok = ByteTokenizer()
    assert tok.detokenize(tok.tokenize("hello")) == "hello"
    ids, types, mask = build_sample(tok, "ab", "cd", 10)
    assert ids.tolist()[:3] == [tok.cls, ord("a"), ord("b")]
    assert ids[3] == tok.sep and ids[6] == tok.sep
    assert types.tolist() == [0, 0, 0, 0, 1, 1, 1, 0, 0, 0]
    assert mask.sum() == 7
    # truncation keeps the pair within budget
    ids, _, _ = build_sample(tok, "x" * 50, "y" * 50, 16)
    assert len(ids) == 16


def test_glue_datasets_parse(tmp_path):
    from tasks.data_utils import ByteTokenizer
    from tasks.glue.data import MNLIDataset, QQPDataset
    mnli = tmp_path / "mnli.tsv"
    qqp = tmp_path / "qqp.tsv"
    _write_mnli(mnli)
    _write_qqp(qqp)
    tok = ByteTokenizer()
    d1 = MNLIDataset("train", [str(mnli)], tok, 64)
    assert len(d1) == 12 and d1[0]["label"].item() == 0
    assert d1[1]["label"].item() == 1
    d2 = QQPDataset("train", [str(qqp)], tok, 64)
    assert len(d2) == 12 and d2[1]["label"].item() == 1
    assert d2[0]["ids"].shape == (64,)


def test_race_dataset_multichoice(tmp_path):
    from tasks.data_utils import ByteTokenizer
    from tasks.race.data import RaceDataset
    f = tmp_path / "race.txt"
    _write_race(f)
    d = RaceDataset("train", [str(f)], ByteTokenizer(), 64)
    assert len(d) == 6
    item = d[2]
    assert item["ids"].shape == (4, 64)
    assert item["label"].item() == 2


def test_wikitext_overlapping_eval_matches_full():
    """With overlapping_eval == seq_length (no overlap) the windowed ppl
    equals a direct full-sequence computation."""
    import torch.nn.functional as F
    from tasks.zeroshot_gpt.evaluate import evaluate_wikitext

    torch.manual_seed(0)
    V = 32

    class Toy(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.emb = torch.nn.Embedding(V, 16)
            self.out = torch.nn.Linear(16, V)

        def forward(self, ids, pos):
            return self.out(self.emb(ids))

    model = Toy().eval()
    tokens = torch.randint(0, V, (33,)).tolist()
    out = evaluate_wikitext(model, "cpu", tokens, seq_length=8,
                            overlapping_eval=8)
    with torch.no_grad():
        ids = torch.tensor(tokens[:-1]).unsqueeze(0)
        logp = F.log_softmax(model(ids, None).float(), -1)[0]
        tgt = torch.tensor(tokens[1:])
        ref = -logp[torch.arange(32), tgt].mean().item()
    import math
    assert abs(out["ppl"] - math.exp(ref)) < 1e-3


def test_tasks_main_mnli_finetune(tmp_path):
    train = tmp_path / "train.tsv"
    dev = tmp_path / "dev.tsv"
    _write_mnli(train)
    _write_mnli(dev)
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29701",
               RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tasks", "main.py"),
         "--task", "MNLI", "--train-data", str(train),
         "--valid-data", str(dev),
         "--num-layers", "2", "--hidden-size", "64",
         "--num-attention-heads", "4", "--seq-length", "32",
         "--max-position-embeddings", "64", "--micro-batch-size", "4",
         "--global-batch-size", "4", "--epochs", "1", "--lr", "1e-4",
         "--hidden-dropout", "0", "--attention-dropout", "0"],
        capture_output=True, text=True, cwd=REPO, env=env, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "validation accuracy" in out.stdout


def test_tasks_main_race_finetune(tmp_path):
    f = tmp_path / "race.txt"
    _write_race(f)
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29702",
               RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tasks", "main.py"),
         "--task", "RACE", "--train-data", str(f), "--valid-data", str(f),
         "--num-layers", "2", "--hidden-size", "64",
         "--num-attention-heads", "4", "--seq-length", "64",
         "--max-position-embeddings", "64", "--micro-batch-size", "2",
         "--global-batch-size", "2", "--epochs", "1", "--lr", "1e-4",
         "--hidden-dropout", "0", "--attention-dropout", "0"],
        capture_output=True, text=True, cwd=REPO, env=env, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "validation accuracy" in out.stdout


def test_tasks_main_zeroshot_eval(tmp_path):
    wiki = tmp_path / "wiki.txt"
    wiki.write_text("the quick brown fox jumps over the lazy dog " * 20)
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29703",
               RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tasks", "main.py"),
         "--task", "WIKITEXT103", "--valid-data", str(wiki),
         "--num-layers", "2", "--hidden-size", "64",
         "--num-attention-heads", "4", "--seq-length", "32",
         "--max-position-embeddings", "64", "--micro-batch-size", "1",
         "--global-batch-size", "1", "--lr", "1e-4",
         "--hidden-dropout", "0", "--attention-dropout", "0"],
        capture_output=True, text=True, cwd=REPO, env=env, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "ppl" in out.stdout


def test_orqa_retriever_eval(tmp_path):
    """Retrieval-accuracy machinery: with k == all blocks every question
    whose answer exists in evidence must be a hit."""
    qa = tmp_path / "qa.jsonl"
    ev = tmp_path / "evidence.jsonl"
    qa.write_text("\n".join(json.dumps(d) for d in [
        {"question": "capital of france", "answers": ["Paris"]},
        {"question": "largest planet", "answers": ["Jupiter"]},
        {"question": "unanswerable", "answers": ["zzz-not-present"]},
    ]))
    ev.write_text("\n".join(json.dumps({"text": t}) for t in [
        "paris is the capital of france",
        "jupiter is the largest planet",
        "unrelated text block",
    ]))
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29704",
               RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tasks", "main.py"),
         "--task", "ORQA", "--qa-data", str(qa),
         "--evidence-data", str(ev),
         "--num-layers", "2", "--hidden-size", "64",
         "--num-attention-heads", "4", "--seq-length", "32",
         "--max-position-embeddings", "64", "--micro-batch-size", "2",
         "--global-batch-size", "2", "--lr", "1e-4",
         "--hidden-dropout", "0", "--attention-dropout", "0"],
        capture_output=True, text=True, cwd=REPO, env=env, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    # top20 covers all 3 blocks: exactly the 2 answerable hit
    assert "top20_accuracy: 66.67%" in out.stdout


def test_msdp_f1_metric():
    from tasks.msdp.metrics import F1Metric
    p, r, f1 = F1Metric.compute_each_pair("the cat sat", "a cat sat down")
    # normalized: {cat, sat} vs {cat, sat, down}
    assert abs(p - 1.0) < 1e-9 and abs(r - 2 / 3) < 1e-9
    assert abs(f1 - 0.8) < 1e-9
    p, r, f1 = F1Metric.compute_all_pairs(["cat"], ["cat"])
    assert f1 == 1.0


def test_msdp_prompt_and_eval(tmp_path):
    inp = tmp_path / "dialog.jsonl"
    inp.write_text("\n".join(json.dumps(d) for d in [
        {"turns": ["hello there", "hi how are you"], "topic": "greeting"},
        {"turns": ["what is rust"], "topic": "programming",
         "knowledge": "rust is a language"},
    ]))
    out = tmp_path / "knwl.txt"
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29706",
               RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tasks", "main.py"),
         "--task", "MSDP-PROMPT", "--prompt-type", "knowledge",
         "--sample-input-file", str(inp),
         "--sample-output-file", str(out),
         "--out-seq-length", "8",
         "--num-layers", "2", "--hidden-size", "64",
         "--num-attention-heads", "4", "--seq-length", "128",
         "--max-position-embeddings", "256", "--micro-batch-size", "1",
         "--global-batch-size", "1", "--lr", "1e-4",
         "--hidden-dropout", "0", "--attention-dropout", "0"],
        capture_output=True, text=True, cwd=REPO, env=env, timeout=420)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = out.read_text().splitlines()
    assert len(lines) == 2
    # F1 eval stage on known files
    guess = tmp_path / "guess.txt"
    ans = tmp_path / "ans.txt"
    guess.write_text("the cat sat\n")
    ans.write_text("a cat sat down\n")
    r2 = subprocess.run(
        [sys.executable, os.path.join(REPO, "tasks", "main.py"),
         "--task", "MSDP-EVAL-F1", "--guess-file", str(guess),
         "--answer-file", str(ans),
         "--num-layers", "2", "--hidden-size", "64",
         "--num-attention-heads", "4", "--seq-length", "32",
         "--max-position-embeddings", "64", "--micro-batch-size", "1",
         "--global-batch-size", "1", "--lr", "1e-4"],
        capture_output=True, text=True, cwd=REPO, env=env, timeout=420)
    assert r2.returncode == 0, r2.stderr[-2000:]
    assert "f1: 0.8000" in r2.stdout
