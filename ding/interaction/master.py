"""HTTP RPC substrate: Master (task dispatcher) and Slave (worker).

Parity: reference ding/interaction/ (master/master.py:25, slave/slave.py:22,
base/network.py:45 HttpEngine): flask servers with heartbeat channels and
POST task dispatch.
"""
import json
import logging
import threading
import time
import uuid
from typing import Any, Callable, Dict, Optional

import requests
from flask import Flask, jsonify, request

logger = logging.getLogger('ding')


def success_response(data: Any = None, message: str = 'success') -> dict:
    return {'success': True, 'code': 0, 'message': message, 'data': data}


def failure_response(code: int = 1, message: str = 'failed', data: Any = None) -> dict:
    return {'success': False, 'code': code, 'message': message, 'data': data}


class HttpEngine:
    """Thin JSON-over-HTTP client."""

    def __init__(self, host: str, port: int, https: bool = False):
        proto = 'https' if https else 'http'
        self.base = f'{proto}://{host}:{port}'

    def request(self, method: str, path: str, data: Optional[dict] = None, timeout: float = 5.0) -> dict:
        url = self.base + path
        resp = requests.request(method, url, json=data, timeout=timeout)
        resp.raise_for_status()
        return resp.json()


class TaskFail:
    """Marker result: the slave declines/fails a task (reference
    ding/interaction/slave/slave.py TaskFail)."""

    def __init__(self, result=None, message: str = ''):
        self.result = result or {}
        self.message = message


class Slave:
    """Worker endpoint: receives tasks over POST /task/new, heartbeats to its
    master. Subclass and override ``_process_task``."""

    def __init__(self, host: str = '127.0.0.1', port: int = 0, heartbeat_span: float = 3.0):
        self._host = host
        self._port = port
        self._heartbeat_span = heartbeat_span
        self._app = Flask(f'ding-slave-{port}')
        self._app.logger.disabled = True
        self._token = None
        self._master: Optional[HttpEngine] = None
        self._current_task = None
        self._task_lock = threading.Lock()
        self._task_result = {}
        self._shutdown = False
        self._register_routes()
        self._server_thread = None
        self._heartbeat_thread = None

    def _register_routes(self):
        app = self._app

        @app.route('/ping', methods=['GET'])
        def ping():
            return jsonify(success_response())

        @app.route('/connect', methods=['POST'])
        def connect():
            body = request.get_json(force=True) or {}
            self._token = body.get('token', uuid.uuid4().hex)
            master_info = body.get('master', {})
            if master_info:
                self._master = HttpEngine(master_info['host'], master_info['port'])
            return jsonify(success_response({'token': self._token}))

        @app.route('/task/new', methods=['POST'])
        def new_task():
            body = request.get_json(force=True) or {}
            task_id = body.get('task_id', uuid.uuid4().hex)
            with self._task_lock:
                if self._current_task is not None:
                    return jsonify(failure_response(code=2, message='busy')), 400
                self._current_task = (task_id, body.get('task', {}))
            threading.Thread(target=self._run_task, args=(task_id, body.get('task', {})), daemon=True).start()
            return jsonify(success_response({'task_id': task_id}))

        @app.route('/task/<task_id>/result', methods=['GET'])
        def task_result(task_id):
            if task_id in self._task_result:
                return jsonify(success_response(self._task_result[task_id]))
            return jsonify(failure_response(code=3, message='pending')), 404

        @app.route('/shutdown', methods=['POST'])
        def shutdown():
            self._shutdown = True
            return jsonify(success_response())

    def _run_task(self, task_id: str, task: dict):
        try:
            result = self._process_task(task)
            self._task_result[task_id] = {'status': 'done', 'result': result}
        except Exception as e:
            self._task_result[task_id] = {'status': 'error', 'error': str(e)}
        finally:
            with self._task_lock:
                self._current_task = None
            if self._master is not None:
                try:
                    self._master.request('POST', '/task/finish', {
                        'task_id': task_id, 'token': self._token, 'result': self._task_result[task_id]
                    })
                except Exception:
                    pass

    def _process_task(self, task: dict) -> Any:
        raise NotImplementedError

    def start(self):
        from werkzeug.serving import make_server
        self._server = make_server(self._host, self._port, self._app, threaded=True)
        self._port = self._server.server_port
        self._server_thread = threading.Thread(target=self._server.serve_forever, daemon=True)
        self._server_thread.start()
        return self

    @property
    def port(self) -> int:
        return self._port

    def close(self):
        self._shutdown = True
        try:
            self._server.shutdown()
        except Exception:
            pass


class Master:
    """Dispatcher: tracks connected slaves, sends tasks, receives results."""

    def __init__(self, host: str = '127.0.0.1', port: int = 0):
        self._host = host
        self._port = port
        self._app = Flask(f'ding-master-{port}')
        self._app.logger.disabled = True
        self._slaves: Dict[str, HttpEngine] = {}
        self._results: Dict[str, dict] = {}
        self._register_routes()

    def _register_routes(self):
        app = self._app

        @app.route('/ping', methods=['GET'])
        def ping():
            return jsonify(success_response())

        @app.route('/task/finish', methods=['POST'])
        def task_finish():
            body = request.get_json(force=True) or {}
            self._results[body['task_id']] = body.get('result')
            return jsonify(success_response())

    def start(self):
        from werkzeug.serving import make_server
        self._server = make_server(self._host, self._port, self._app, threaded=True)
        self._port = self._server.server_port
        self._thread = threading.Thread(target=self._server.serve_forever, daemon=True)
        self._thread.start()
        return self

    @property
    def port(self) -> int:
        return self._port

    def connect_slave(self, name: str, host: str, port: int) -> None:
        engine = HttpEngine(host, port)
        engine.request('POST', '/connect', {
            'token': uuid.uuid4().hex, 'master': {'host': self._host, 'port': self._port}
        })
        self._slaves[name] = engine

    def new_task(self, name: str, task: dict) -> str:
        task_id = uuid.uuid4().hex
        self._slaves[name].request('POST', '/task/new', {'task_id': task_id, 'task': task})
        return task_id

    def wait_task(self, task_id: str, timeout: float = 30.0) -> dict:
        start = time.time()
        while time.time() - start < timeout:
            if task_id in self._results:
                return self._results[task_id]
            time.sleep(0.05)
        raise TimeoutError(f'task {task_id} result timeout')

    def close(self):
        for name, engine in self._slaves.items():
            try:
                engine.request('POST', '/shutdown')
            except Exception:
                pass
        try:
            self._server.shutdown()
        except Exception:
            pass
