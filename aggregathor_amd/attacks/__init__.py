"""Byzantine gradient-attack plugin layer.

The reference parses ``--attack`` / ``--attack-args`` / ``--nb-real-byz-
workers`` but leaves the wiring as a TODO (/root/reference/runner.py:149-155,
345); its actual attacks are the ``mnistAttack`` data-poisoning experiment
and the lossy UDP transport. This package wires the flags for real: an
attack transforms the gradient a real-Byzantine worker submits.

Convention: the real Byzantine workers are worker ids ``0 .. nb_real-1``
(consistent with ``mnistAttack`` poisoning worker 0). Each Byzantine worker
still computes its honest gradient, then ``craft`` replaces it -- so attacks
that depend on an honest estimate (e.g. gradient reversal) work without any
extra communication on any rank placement.
"""

import os

from .. import tools


class _Attack:
    """Base class of all gradient attacks."""

    def __init__(self, args):
        raise NotImplementedError

    def craft(self, honest, worker, step):
        """Return the Byzantine gradient submitted instead of ``honest``.

        Args:
          honest: [d] tensor, the worker's honestly-computed flat gradient
                  (must not be modified in place).
          worker: global worker id (one of the real-Byzantine ids).
          step:   global step number.
        Returns:
          [d] tensor on the same device/dtype.
        """
        raise NotImplementedError


_register = tools.ClassRegister("attack")


def itemize():
    return _register.itemize()


def register(name, cls):
    return _register.register(name, cls)


def instantiate(name, args=None):
    return _register.instantiate(name, args or [])


tools.import_directory(__name__, os.path.dirname(__file__))
