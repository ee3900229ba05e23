"""Built-in connectors for commonly used systems.

Parity target: ``bytewax.connectors`` — files, CSV, stdio, demo, Kafka.
"""
