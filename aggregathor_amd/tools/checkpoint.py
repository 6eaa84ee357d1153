"""Checkpoint manager: ``model-<step>`` file layout with restore-latest.

Re-expression of the reference's ``tools.Checkpoints``
(/root/reference/tools/tf.py:78-173) for PyTorch state dicts: checkpoints are
``<dir>/<base>-<step>.ckpt`` files, discovery sorts by step, ``restore``
loads the latest by default, and ``get()`` returns newly-appeared files once
(used by the evaluation thread to follow a training run from another
process).
"""

import pathlib

import torch


class Checkpoints:
    """Simple checkpoint manager with the reference's file-layout semantics."""

    SUFFIX = ".ckpt"

    def __init__(self, path, base=None):
        from .. import config
        self.__path = pathlib.Path(path)
        self.__base = base if base is not None else config.default_checkpoint_base_name
        self.__available = []
        self.__processed = set()

    def _update(self):
        available = []
        if self.__path.exists():
            for item in self.__path.iterdir():
                if (item.is_file() and item.suffix == self.SUFFIX
                        and item.stem.startswith(self.__base + "-")):
                    available.append(item)
            available.sort(key=lambda p: int(p.stem[p.stem.rindex("-") + 1:]))
        self.__available = available

    def get(self, no_filter=False):
        """List available checkpoint files, excluding previously-returned ones."""
        self._update()
        if no_filter:
            got = list(self.__available)
        else:
            got = [e for e in self.__available if e not in self.__processed]
        self.__processed.update(got)
        return got

    def can_restore(self):
        self._update()
        return len(self.__available) > 0

    @staticmethod
    def step_of(path):
        """Extract the global step encoded in a checkpoint file name."""
        stem = pathlib.Path(path).stem
        return int(stem[stem.rindex("-") + 1:])

    def restore(self, path=None, map_location="cpu"):
        """Load a checkpoint payload (latest if ``path`` is None).

        ``weights_only=True``: Trainer auto-restores the latest file at
        startup, so an unpickle here would turn a tampered checkpoint dir
        into code execution. Payloads are plain dicts of tensors/ints
        (Engine.state_dict), which the weights-only unpickler accepts.
        """
        from . import UserException
        self._update()
        if path is None:
            if not self.__available:
                raise UserException("No storage file to restore")
            path = self.__available[-1]
        return torch.load(str(path), map_location=map_location, weights_only=True)

    def save(self, payload, step):
        """Atomically save a checkpoint payload at the given step."""
        self.__path.mkdir(parents=True, exist_ok=True)
        final = self.__path / f"{self.__base}-{step}{self.SUFFIX}"
        tmp = final.with_suffix(final.suffix + ".tmp")
        torch.save(payload, str(tmp))
        tmp.replace(final)
        return final
