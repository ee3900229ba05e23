"""Drop-in compatibility aliases for Bytewax user code.

``import bytewax.operators as op`` (and every other public Bytewax
module path) resolves to the corresponding :mod:`bytewax_amd` module,
so dataflow definitions written against the reference API run
unchanged on this framework:

```python
import bytewax.operators as op
from bytewax.dataflow import Dataflow
from bytewax.testing import TestingSource, run_main
```
"""

import importlib
import sys

_ALIASES = [
    "dataflow",
    "errors",
    "inputs",
    "outputs",
    "operators",
    "operators.windowing",
    "operators.helpers",
    "connectors",
    "connectors.files",
    "connectors.stdio",
    "connectors.demo",
    "connectors.kafka",
    "connectors.kafka.operators",
    "connectors.kafka.serde",
    "recovery",
    "run",
    "testing",
    "tracing",
    "visualize",
    "_metrics",
    "_utils",
]

for _name in _ALIASES:
    _mod = importlib.import_module(f"bytewax_amd.{_name}")
    sys.modules[f"bytewax.{_name}"] = _mod
    # Attach top-level attributes (``bytewax.operators``) so attribute
    # access works as well as direct imports.
    if "." not in _name:
        globals()[_name] = _mod

__version__ = importlib.import_module("bytewax_amd").__version__
