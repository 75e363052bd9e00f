"""Generic finetune/eval loop for downstream tasks (reference
tasks/finetune_utils.py:1-306).

``ClassificationModel`` is a BERT tower pooled at [CLS] with a
dense→tanh→dense head; ``finetune`` runs epoch-based training with
per-epoch validation accuracy (reference's finetune()/_train loop),
data-parallel when torch.distributed is initialized.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from megatronapp_amd.core import parallel_state
from megatronapp_amd.core.models.bert import BertModel
from megatronapp_amd.core.models.bert.bert_layer_specs import (
    get_bert_layer_local_spec,
)
from megatronapp_amd.core.transformer.module import MegatronModule
from megatronapp_amd.training.global_vars import get_args


class ClassificationModel(MegatronModule):
    """BERT + [CLS] classification head (reference
    legacy/model/classification.py)."""

    def __init__(self, config, vocab_size, max_sequence_length,
                 num_classes: int):
        super().__init__(config=config)
        self.num_classes = num_classes
        self.language_model = BertModel(
            config=config,
            transformer_layer_spec=get_bert_layer_local_spec(),
            vocab_size=vocab_size,
            max_sequence_length=max_sequence_length,
            add_binary_head=False, post_process=False)
        h = config.hidden_size
        self.pool_dense = torch.nn.Linear(h, h)
        self.head = torch.nn.Linear(h, num_classes)

    def set_input_tensor(self, input_tensor):
        pass

    def forward(self, input_ids, attention_mask, tokentype_ids):
        hidden = self.language_model(
            input_ids, attention_mask, tokentype_ids=tokentype_ids)
        pooled = torch.tanh(self.pool_dense(hidden[0]))   # [b, h]
        return self.head(pooled)


def _loader(dataset, batch_size, shuffle):
    sampler = None
    if torch.distributed.is_initialized() and \
            torch.distributed.get_world_size() > 1:
        sampler = torch.utils.data.distributed.DistributedSampler(
            dataset, shuffle=shuffle)
        shuffle = False
    return torch.utils.data.DataLoader(
        dataset, batch_size=batch_size, shuffle=shuffle, sampler=sampler,
        drop_last=False)


def _score(model, batch, device):
    """Returns [b, classes] logits; multichoice [b, k, s] inputs
    collapse into the batch dimension and score one logit per choice
    (reference race/finetune.py sample_multiplier handling)."""
    ids = batch["ids"].to(device)
    mask = batch["mask"].to(device)
    types = batch["types"].to(device)
    if ids.dim() == 3:
        b, k, s = ids.shape
        logits = model(ids.view(b * k, s), mask.view(b * k, s),
                       types.view(b * k, s))
        return logits.view(b, k)
    return model(ids, mask, types)


def accuracy(model, dataset, batch_size, device) -> float:
    model.eval()
    correct = total = 0
    with torch.no_grad():
        for batch in _loader(dataset, batch_size, shuffle=False):
            logits = _score(model, batch, device)
            pred = logits.argmax(-1)
            correct += (pred == batch["label"].to(device)).sum().item()
            total += len(pred)
    model.train()
    if torch.distributed.is_initialized() and \
            torch.distributed.get_world_size() > 1:
        t = torch.tensor([correct, total], dtype=torch.float64)
        torch.distributed.all_reduce(t)
        correct, total = t.tolist()
    return correct / max(total, 1)


def finetune(train_ds, valid_ds, num_classes: int, name: str = "task"):
    """Epoch-based finetune with per-epoch validation accuracy.
    Returns the final validation accuracy."""
    args = get_args()
    from megatronapp_amd.training.arguments import (
        core_transformer_config_from_args)
    config = core_transformer_config_from_args(args)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    model = ClassificationModel(
        config, args.padded_vocab_size, args.max_position_embeddings,
        num_classes).to(device)
    if torch.distributed.is_initialized() and \
            torch.distributed.get_world_size() > 1:
        model = torch.nn.parallel.DistributedDataParallel(model)
    opt = torch.optim.AdamW(model.parameters(), lr=args.lr,
                            weight_decay=args.weight_decay)
    epochs = args.epochs if args.epochs is not None else 3
    rank0 = (not torch.distributed.is_initialized()
             or torch.distributed.get_rank() == 0)
    acc = accuracy(model, valid_ds, args.micro_batch_size, device) \
        if epochs == 0 else 0.0
    for epoch in range(epochs):
        loader = _loader(train_ds, args.micro_batch_size, shuffle=True)
        if hasattr(loader.sampler, "set_epoch"):
            loader.sampler.set_epoch(epoch)
        for it, batch in enumerate(loader):
            logits = _score(model, batch, device)
            loss = F.cross_entropy(logits, batch["label"].to(device))
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
            if rank0 and it % max(1, args.log_interval) == 0:
                print(f"{name} epoch {epoch} iter {it} "
                      f"lm loss: {loss.item():.4f}", flush=True)
        acc = accuracy(model, valid_ds, args.micro_batch_size, device)
        if rank0:
            print(f"{name} epoch {epoch} validation accuracy: "
                  f"{acc * 100:.2f}%", flush=True)
    return acc
