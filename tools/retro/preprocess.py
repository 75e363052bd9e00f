#!/usr/bin/env python3
"""Retro preprocessing pipeline (reference tools/retro +
core/datasets/retro): chunk database build, chunk embedding, neighbor
search, and project-directory export.

The reference builds a faiss index; faiss is not in this image, so the
index is brute-force maximum-inner-product over normalized embeddings,
computed blockwise — on MI355X each block is one hipBLASLt GEMM + topk,
which is exact (no ANN approximation) and fast enough for
moderate-sized chunk databases.

Project layout written by ``build_retro_project``:
  chunks.npy          [n_chunks, chunk_length] int32 token chunks
  chunk_doc.npy       [n_chunks] document id per chunk
  neighbors.npy       [n_chunks, k] neighbor chunk ids (same-doc excluded)
  config.json         chunk_length / k / num_retrieved_chunks

``load_neighbor_tokens`` resolves neighbor ids to neighbor+continuation
token windows of ``retrieved_length`` at training time.
"""

from __future__ import annotations

import json
import os

import numpy as np
import torch


def build_chunk_db(token_docs, chunk_length: int, pad_id: int):
    """List of per-document token arrays -> (chunks [n, m], doc_ids [n]).
    The tail of each document pads to a full chunk (reference chunk-db
    semantics: chunks never span documents)."""
    chunks, doc_ids = [], []
    for d, toks in enumerate(token_docs):
        toks = np.asarray(toks, dtype=np.int32)
        for s in range(0, len(toks), chunk_length):
            c = toks[s:s + chunk_length]
            if len(c) < chunk_length:
                c = np.concatenate(
                    [c, np.full(chunk_length - len(c), pad_id,
                                dtype=np.int32)])
            chunks.append(c)
            doc_ids.append(d)
    return np.stack(chunks), np.asarray(doc_ids, dtype=np.int64)


class BruteForceMIPSIndex:
    """Exact maximum-inner-product search over normalized embeddings."""

    def __init__(self, embeddings: np.ndarray, device: str = None):
        self.device = device or (
            "cuda" if torch.cuda.is_available() else "cpu")
        e = torch.as_tensor(embeddings, dtype=torch.float32,
                            device=self.device)
        self.base = torch.nn.functional.normalize(e, dim=1)

    def search(self, queries: np.ndarray, k: int,
               query_docs: np.ndarray = None,
               base_docs: np.ndarray = None,
               block: int = 4096):
        """Top-k ids per query; same-document hits are excluded when the
        doc maps are given (the reference excludes the query's own
        document so training neighbors are non-trivial)."""
        q = torch.nn.functional.normalize(
            torch.as_tensor(queries, dtype=torch.float32,
                            device=self.device), dim=1)
        bd = None if base_docs is None else torch.as_tensor(
            base_docs, device=self.device)
        out = []
        for s in range(0, q.shape[0], block):
            qb = q[s:s + block]
            scores = qb @ self.base.t()                 # [b, N]
            if bd is not None and query_docs is not None:
                qd = torch.as_tensor(query_docs[s:s + block],
                                     device=self.device)
                scores.masked_fill_(qd.unsqueeze(1) == bd.unsqueeze(0),
                                    float("-inf"))
            out.append(scores.topk(k, dim=1).indices.cpu())
        return torch.cat(out).numpy()


def build_retro_project(out_dir: str, token_docs, embedder, pad_id: int,
                        chunk_length: int = 64, num_neighbors: int = 2,
                        num_retrieved_chunks: int = 2):
    """Chunk -> embed -> search -> write the project directory."""
    os.makedirs(out_dir, exist_ok=True)
    chunks, doc_ids = build_chunk_db(token_docs, chunk_length, pad_id)
    emb = embedder.embed_tokens(chunks, pad_id)
    index = BruteForceMIPSIndex(emb)
    k = min(num_neighbors, len(chunks) - 1)
    neighbors = index.search(emb, k, query_docs=doc_ids,
                             base_docs=doc_ids)
    np.save(os.path.join(out_dir, "chunks.npy"), chunks)
    np.save(os.path.join(out_dir, "chunk_doc.npy"), doc_ids)
    np.save(os.path.join(out_dir, "neighbors.npy"), neighbors)
    with open(os.path.join(out_dir, "config.json"), "w") as f:
        json.dump({"chunk_length": chunk_length,
                   "num_neighbors": num_neighbors,
                   "num_retrieved_chunks": num_retrieved_chunks,
                   "pad_id": pad_id}, f)
    return chunks, doc_ids, neighbors


def load_retro_project(project_dir: str):
    chunks = np.load(os.path.join(project_dir, "chunks.npy"))
    doc_ids = np.load(os.path.join(project_dir, "chunk_doc.npy"))
    neighbors = np.load(os.path.join(project_dir, "neighbors.npy"))
    with open(os.path.join(project_dir, "config.json")) as f:
        cfg = json.load(f)
    return chunks, doc_ids, neighbors, cfg


def load_neighbor_tokens(chunks: np.ndarray, doc_ids: np.ndarray,
                         neighbor_ids: np.ndarray, pad_id: int,
                         num_retrieved_chunks: int = 2) -> np.ndarray:
    """[k] neighbor chunk ids -> [k, r] neighbor+continuation tokens:
    each retrieved sequence is the neighbor chunk followed by its
    same-document continuation chunks (reference retro_retrieved_length
    = num_retrieved_chunks * chunk_length)."""
    m = chunks.shape[1]
    r = num_retrieved_chunks * m
    out = np.full((len(neighbor_ids), r), pad_id, dtype=np.int64)
    for j, cid in enumerate(neighbor_ids):
        parts = [chunks[cid]]
        for nxt in range(1, num_retrieved_chunks):
            cand = cid + nxt
            if cand < len(chunks) and doc_ids[cand] == doc_ids[cid]:
                parts.append(chunks[cand])
            else:
                break
        seq = np.concatenate(parts)
        out[j, :len(seq)] = seq
    return out
