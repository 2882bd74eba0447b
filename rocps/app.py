"""``rocps.app.run()`` — the reference's ``tf.app.run()`` entry helper."""

from __future__ import annotations

import sys
from typing import Callable, List, Optional

from rocps.config import FLAGS


def run(main: Optional[Callable] = None, argv: Optional[List[str]] = None) -> None:
    """Parse flags, then call ``main(argv)`` and exit with its return value."""
    if main is None:
        main = sys.modules["__main__"].main  # type: ignore[attr-defined]
    rest = FLAGS._parse(sys.argv[1:] if argv is None else argv[1:])
    ret = main([sys.argv[0]] + rest)
    sys.exit(ret if isinstance(ret, int) else 0)
