"""MSDP F1 evaluation (reference tasks/msdp/evaluate.py)."""

from .metrics import F1Metric


def evaluate_f1(guess_file: str, answer_file: str):
    def read(path, drop=("<|endoftext|>",), empty=("no_passages_used",)):
        out = []
        with open(path, encoding="utf-8") as f:
            for line in f:
                line = line.strip()
                for d in drop:
                    line = line.replace(d, "")
                if line in empty:
                    line = ""
                out.append(line)
        return out

    guesses = read(guess_file)
    answers = read(answer_file)
    assert len(guesses) == len(answers), \
        "lengths of guess and answer are different!"
    p, r, f1 = F1Metric.compute_all_pairs(guesses, answers)
    print(f"Precision: {p:.4f}; recall: {r:.4f}; f1: {f1:.4f}",
          flush=True)
    return p, r, f1
