"""Gunicorn configuration (reference gunicorn_config.py:14-125 analog).

IMPORTANT ARCHITECTURAL DIFFERENCE: the reference ran (2*CPU+1) worker
processes, each with its OWN SwarmsDB state and only Kafka shared
(SURVEY.md §8.8). Here the GPU queue is the shared state and exactly ONE
worker process owns the device; request concurrency comes from the async
event loop, throughput from the GPU. Scaling beyond one process means
one process per GPU with agent sharding (swarmdb_amd/parallel), not
N workers on one queue.
"""

import os

bind = f"0.0.0.0:{os.environ.get('PORT', '8000')}"
workers = 1  # single device-owner (see module docstring)
worker_class = "uvicorn.workers.UvicornWorker"

# worker recycling kept from the reference (gunicorn_config.py:38-41)
max_requests = int(os.environ.get("GUNICORN_MAX_REQUESTS", "10000"))
max_requests_jitter = int(os.environ.get("GUNICORN_MAX_REQUESTS_JITTER", "1000"))

timeout = int(os.environ.get("GUNICORN_TIMEOUT", "120"))
graceful_timeout = int(os.environ.get("GUNICORN_GRACEFUL_TIMEOUT", "60"))
keepalive = int(os.environ.get("GUNICORN_KEEPALIVE", "5"))

accesslog = os.environ.get("GUNICORN_ACCESS_LOG", "-")
errorlog = os.environ.get("GUNICORN_ERROR_LOG", "-")
loglevel = os.environ.get("GUNICORN_LOG_LEVEL", "info")
access_log_format = '%(h)s %(l)s %(u)s %(t)s "%(r)s" %(s)s %(b)s "%(f)s" "%(a)s" %(L)s'

# usage: gunicorn -c deploy/gunicorn_config.py 'swarmdb_amd.api.app:get_app()'
