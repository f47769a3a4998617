"""Minimal @function app.

    b9 run examples/hello_function.py:hello --args '{"x": 21}'
"""

import beta9_amd


@beta9_amd.function(cpu=1000, memory=512)
def hello(x: int = 1):
    return {"answer": x * 2}
