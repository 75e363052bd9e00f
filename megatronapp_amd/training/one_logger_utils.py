"""E2E metrics hooks (reference training/one_logger_utils.py wraps the
proprietary one_logger service).  That service does not exist here; the
same call surface aggregates into an in-process dict so callers and tests
keep working, and `get_metrics()` exposes what would have been shipped."""

import time

_METRICS = {}


def on_pretrain_start():
    _METRICS["app_start_time"] = time.time()


def on_train_start(iteration=0, consumed_train_samples=0, **kwargs):
    _METRICS.update(train_start_time=time.time(),
                    start_iteration=iteration,
                    start_consumed_samples=consumed_train_samples)


def track_e2e_metrics(**kwargs):
    _METRICS.update(kwargs)


def on_save_checkpoint_start(*a, **k):
    _METRICS["last_ckpt_start"] = time.time()


def on_save_checkpoint_success(*a, **k):
    _METRICS["last_ckpt_seconds"] = time.time() - _METRICS.get(
        "last_ckpt_start", time.time())


def on_train_end(*a, **k):
    _METRICS["train_end_time"] = time.time()


def get_metrics():
    return dict(_METRICS)
