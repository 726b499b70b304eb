"""FedCallHolder — the heart of multi-controller execution.

Parity: /root/reference/fed/_private/fed_call_holder.py:62-110.  Every party
executes the same driver line; the holder decides, per party:

- **this party owns the task** → resolve FedObject args to local refs (recv
  barriers for foreign ones), submit the real task to the in-party executor,
  return FedObjects wrapping live ObjectRefs;
- **another party owns it** → flatten the args, push every *locally owned*
  FedObject arg that was not already sent to that party (dedup via the
  object's sending context), and return data-less FedObjects.

The allocated ``fed_task_id`` comes from the deterministic seq-id counter, so
both parties name the same call identically without negotiation.
"""
from __future__ import annotations

import logging
from typing import Callable, Optional

from rayfed_amd import tree_util
from rayfed_amd._private.global_context import get_global_context
from rayfed_amd.fed_object import FedObject

logger = logging.getLogger(__name__)


class FedCallHolder:
    def __init__(
        self,
        node_party: str,
        submit_fn: Callable,
        options: Optional[dict] = None,
    ):
        """``submit_fn(resolved_args, resolved_kwargs)`` submits the real task
        locally and returns an ObjectRef (or a list for num_returns>1)."""
        self._node_party = node_party
        self._submit_fn = submit_fn
        self._options = options or {}

    def options(self, **options):
        self._options = options
        return self

    def internal_remote(self, *args, **kwargs):
        ctx = get_global_context()
        if ctx is None:
            raise RuntimeError("fed.init must be called before fed calls")
        current_party = ctx.get_current_party()
        fed_task_id = ctx.next_seq_id()
        num_returns = self._options.get("num_returns", 1)

        if current_party == self._node_party:
            from rayfed_amd.utils import resolve_dependencies

            resolved_args, resolved_kwargs = resolve_dependencies(
                current_party, fed_task_id, *args, **kwargs
            )
            ref = self._submit_fn(resolved_args, resolved_kwargs)
            if num_returns == 1:
                return FedObject(self._node_party, fed_task_id, ref)
            return [
                FedObject(self._node_party, fed_task_id, sub_ref, i)
                for i, sub_ref in enumerate(ref)
            ]

        # Another party executes this call: push our owned args it needs.
        flattened, _ = tree_util.tree_flatten((args, kwargs))
        for arg in flattened:
            if not isinstance(arg, FedObject):
                continue
            if arg.get_party() != current_party:
                continue
            if arg.was_sending_or_sent_to_party(self._node_party):
                logger.debug(
                    "%s already sent to %s; skipping dup send",
                    arg.get_fed_task_id(),
                    self._node_party,
                )
                continue
            from rayfed_amd.proxy.barriers import send

            arg.mark_is_sending_to_party(self._node_party)
            send(
                dest_party=self._node_party,
                data=arg.get_ray_object_ref(),
                upstream_seq_id=arg.get_fed_task_id(),
                downstream_seq_id=fed_task_id,
            )
        if num_returns == 1:
            return FedObject(self._node_party, fed_task_id, None)
        return [
            FedObject(self._node_party, fed_task_id, None, i)
            for i in range(num_returns)
        ]
