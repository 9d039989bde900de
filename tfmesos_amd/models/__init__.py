"""Model families ported from the reference examples:

* ``mlp``   — the mnist_replica 784-100-10 MLP (the benchmark workload,
  reference ``examples/mnist/mnist_replica.py:116-145``)
* ``nmf``   — rank-200 nonnegative matrix factorization (sparse-model
  stand-in, reference ``examples/matrix_factorization.py``)
* ``inception`` — Inception-v3-class conv net (BASELINE.json conv config)
"""
