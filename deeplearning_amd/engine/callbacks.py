"""Callbacks: named training-lifecycle hook registry.

Reference parity: detection/yolov5/utils/callbacks.py:8-66 (Callbacks class
with register_action/run) — same surface, plus decorator registration.
"""
from __future__ import annotations


HOOKS = (
    "on_pretrain_routine_start", "on_pretrain_routine_end",
    "on_train_start", "on_train_epoch_start", "on_train_batch_start",
    "on_train_batch_end", "on_train_epoch_end",
    "on_val_start", "on_val_batch_start", "on_val_batch_end", "on_val_end",
    "on_fit_epoch_end", "on_model_save", "on_train_end",
)


class Callbacks:
    def __init__(self):
        self._callbacks = {h: [] for h in HOOKS}

    def register_action(self, hook: str, name: str = "", callback=None):
        assert hook in self._callbacks, \
            f"hook '{hook}' not in {sorted(self._callbacks)}"
        assert callable(callback), f"callback '{callback}' is not callable"
        self._callbacks[hook].append({"name": name, "callback": callback})

    def on(self, hook: str, name: str = ""):
        """Decorator form: @callbacks.on('on_train_epoch_end')."""
        def deco(fn):
            self.register_action(hook, name or fn.__name__, fn)
            return fn
        return deco

    def get_registered_actions(self, hook: str | None = None):
        return self._callbacks[hook] if hook else self._callbacks

    def run(self, hook: str, *args, **kwargs):
        assert hook in self._callbacks, f"unknown hook '{hook}'"
        for entry in self._callbacks[hook]:
            entry["callback"](*args, **kwargs)
