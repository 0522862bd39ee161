"""Runnable middleware-pipeline example mains.

Parity: reference ding/example/*.py (~40 single-file mains composing the
Task runtime). Each example here is a `main(max_step=...)` function over the
self-contained dizoo envs so it runs offline; invoke as
`python -m ding.example.dqn` etc.
"""
