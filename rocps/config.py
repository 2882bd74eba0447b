"""Flag system mirroring the reference's ``tf.app.flags`` surface.

The reference scripts define flags like::

    tf.app.flags.DEFINE_string("ps_hosts", "", "...")
    FLAGS = tf.app.flags.FLAGS

so reference run commands (``--ps_hosts=... --worker_hosts=...
--job_name=worker --task_index=0 --issync=1``) parse unchanged here.
Backed by argparse; supports ``--flag=value`` and ``--flag value``.
"""

from __future__ import annotations

import argparse
import sys
from typing import Any, Dict, List, Optional


class _FlagValues:
    """The FLAGS singleton. Attribute access triggers parsing (like absl)."""

    def __init__(self) -> None:
        object.__setattr__(self, "_defs", {})  # name -> (type, default, help)
        object.__setattr__(self, "_values", {})
        object.__setattr__(self, "_parsed", False)
        object.__setattr__(self, "_unparsed", [])

    # -- definition ---------------------------------------------------------
    def _define(self, name: str, default: Any, help_: str, type_: type) -> None:
        self._defs[name] = (type_, default, help_)
        self._values[name] = default

    # -- parsing ------------------------------------------------------------
    def _parse(self, argv: Optional[List[str]] = None) -> List[str]:
        argv = list(sys.argv[1:] if argv is None else argv)
        parser = argparse.ArgumentParser(add_help=False, allow_abbrev=False)
        for name, (type_, default, help_) in self._defs.items():
            if type_ is bool:
                parser.add_argument(
                    "--" + name, type=_parse_bool, default=default,
                    nargs="?", const=True, help=help_)
            else:
                parser.add_argument("--" + name, type=type_, default=default,
                                    help=help_)
        ns, rest = parser.parse_known_args(argv)
        for name in self._defs:
            self._values[name] = getattr(ns, name)
        object.__setattr__(self, "_parsed", True)
        object.__setattr__(self, "_unparsed", rest)
        return rest

    def mark_as_parsed(self) -> None:
        object.__setattr__(self, "_parsed", True)

    # -- access -------------------------------------------------------------
    def __getattr__(self, name: str) -> Any:
        defs = object.__getattribute__(self, "_defs")
        if name.startswith("_") or name not in defs:
            raise AttributeError(name)
        if not object.__getattribute__(self, "_parsed"):
            self._parse()
        return object.__getattribute__(self, "_values")[name]

    def __setattr__(self, name: str, value: Any) -> None:
        if name in self._defs:
            self._values[name] = value
        else:
            object.__setattr__(self, name, value)

    def __contains__(self, name: str) -> bool:
        return name in self._defs

    def flag_values_dict(self) -> Dict[str, Any]:
        if not self._parsed:
            self._parse()
        return dict(self._values)


def _parse_bool(v: Any) -> bool:
    if isinstance(v, bool):
        return v
    s = str(v).lower()
    if s in ("1", "true", "t", "yes", "y"):
        return True
    if s in ("0", "false", "f", "no", "n", ""):
        return False
    raise argparse.ArgumentTypeError(f"invalid bool: {v!r}")


FLAGS = _FlagValues()


def DEFINE_string(name: str, default: Optional[str], help: str = "") -> None:  # noqa: A002
    FLAGS._define(name, default, help, str)


def DEFINE_integer(name: str, default: Optional[int], help: str = "") -> None:  # noqa: A002
    FLAGS._define(name, default, help, int)


def DEFINE_float(name: str, default: Optional[float], help: str = "") -> None:  # noqa: A002
    FLAGS._define(name, default, help, float)


def DEFINE_boolean(name: str, default: Optional[bool], help: str = "") -> None:  # noqa: A002
    FLAGS._define(name, default, help, bool)


DEFINE_bool = DEFINE_boolean
