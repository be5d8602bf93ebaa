"""Evaluation: match executors, in-training evaluator, offline evaluation
pool, network battle mode and ONNX inference.

Protocol contracts kept from the reference (evaluation.py): the
update/action/observe/outcome RPC grammar with string-encoded actions and
``diff_info`` replica sync, the match-server port (:9876, overridable),
the CLI argument meanings of the --eval/--eval-server/--eval-client modes
and the per-pattern win-rate report.  The implementations are this repo's
own — table-dispatched RPC handling, a job-list + result-table evaluation
pool, and a restructured ONNX session wrapper.
"""

import multiprocessing as mp
import random
import time

from .environment import prepare_env, make_env
from .connection import send_recv, accept_socket_connections, \
    connect_socket_connection
from .agent import RandomAgent, RuleBasedAgent, Agent, EnsembleAgent, SoftAgent

import os as _os
NETWORK_MATCH_PORT = int(_os.environ.get('HANDYRL_MATCH_PORT', 9876))


def view(env, player=None):
    if hasattr(env, 'view'):
        env.view(player=player)
    else:
        print(env)


def view_transition(env):
    if hasattr(env, 'view_transition'):
        env.view_transition()


# -- network battle RPC ------------------------------------------------------

class NetworkAgentClient:
    """Client side of a network battle: holds the local agent and a replica
    environment, executes RPC commands streamed by the server until 'quit'
    or disconnect."""

    def __init__(self, agent, env, conn):
        self.conn = conn
        self.agent = agent
        self.env = env

    # each handler returns the RPC reply
    def _rpc_action(self, player):
        view(self.env)
        chosen = self.agent.action(self.env, player, show=True)
        return self.env.action2str(chosen, player)

    def _rpc_observe(self, player):
        view(self.env)
        return self.agent.observe(self.env, player, show=True)

    def _rpc_update(self, info, reset):
        reply = self.env.update(info, reset)
        if reset:
            self.agent.reset(self.env, show=True)
        else:
            view_transition(self.env)
        return reply

    def run(self):
        handlers = {'action': self._rpc_action, 'observe': self._rpc_observe,
                    'update': self._rpc_update}
        while True:
            try:
                command, args = self.conn.recv()
            except (ConnectionResetError, EOFError):
                break
            if command == 'quit':
                break
            if command == 'outcome':
                print('outcome = %f' % args[0])
                self.conn.send(None)
                continue
            handler = handlers.get(command)
            if handler is not None:
                self.conn.send(handler(*args))
            else:                       # any other env method, e.g. reset
                self.conn.send(getattr(self.env, command)(*args))


class NetworkAgent:
    """Server-side proxy for one remote client: every agent call becomes
    one RPC round trip."""

    def __init__(self, conn):
        self.conn = conn

    def _call(self, command, *args):
        return send_recv(self.conn, (command, list(args)))

    def update(self, data, reset):
        return self._call('update', data, reset)

    def outcome(self, outcome):
        return self._call('outcome', outcome)

    def action(self, player):
        return self._call('action', player)

    def observe(self, player):
        return self._call('observe', player)


# -- match executors ---------------------------------------------------------

def _collect_actions(env, agents, show):
    """One step's actions from every turn player; observers observe."""
    turn_players, observers = env.turns(), env.observers()
    actions = {}
    for p, agent in agents.items():
        if p in turn_players:
            actions[p] = agent.action(env, p, show=show)
        elif p in observers:
            agent.observe(env, p, show=show)
    return actions


def exec_match(env, agents, critic=None, show=False, game_args={}):
    """Match over one shared environment instance."""
    if env.reset(game_args):
        return None
    for agent in agents.values():
        agent.reset(env, show=show)
    while not env.terminal():
        if show:
            view(env)
            if critic is not None:
                print('cv = ', critic.observe(env, None, show=False)[0])
        if env.step(_collect_actions(env, agents, show)):
            return None
        if show:
            view_transition(env)
    outcome = env.outcome()
    if show:
        print('final outcome = %s' % outcome)
    return {'result': outcome}


def exec_network_match(env, network_agents, critic=None, show=False,
                       game_args={}):
    """Match where each remote side holds its own replica, synced only by
    diff_info/update strings (partial information stays partial)."""
    if env.reset(game_args):
        return None
    for p, agent in network_agents.items():
        agent.update(env.diff_info(p), True)
    while not env.terminal():
        if show:
            view(env)
            if critic is not None:
                print('cv = ', critic.observe(env, None, show=False)[0])
        turn_players, observers = env.turns(), env.observers()
        actions = {}
        for p, agent in network_agents.items():
            if p in turn_players:
                actions[p] = env.str2action(agent.action(p), p)
            elif p in observers:
                agent.observe(p)
        if env.step(actions):
            return None
        for p, agent in network_agents.items():
            agent.update(env.diff_info(p), False)
    outcome = env.outcome()
    for p, agent in network_agents.items():
        agent.outcome(outcome[p])
    return {'result': outcome}


# -- in-training evaluator ---------------------------------------------------

def build_agent(raw, env=None):
    """Opponent spec -> agent, for specs that need no model file."""
    if raw == 'random':
        return RandomAgent()
    if isinstance(raw, str) and raw.startswith('rulebase'):
        _, _, key = raw.partition('-')
        return RuleBasedAgent(key or None)
    return None


class Evaluator:
    """In-training evaluation job: the latest model occupies the seats
    named in the job; a configured opponent fills the rest."""

    def __init__(self, env, args):
        self.env = env
        self.args = args
        self.default_opponent = 'random'

    def execute(self, models, args):
        choices = self.args.get('eval', {}).get('opponent', [])
        opponent = random.choice(choices) if choices else self.default_opponent
        agents = {p: Agent(m) if m is not None else build_agent(opponent, self.env)
                  for p, m in models.items()}
        results = exec_match(self.env, agents)
        if results is None:
            print('None episode in evaluation!')
            return None
        return {'args': args, 'opponent': opponent, **results}


# -- offline evaluation pool -------------------------------------------------

def wp_func(results):
    """{outcome: count} -> win probability ((mean+1)/2 convention)."""
    games = sum(n for oc, n in results.items() if oc is not None)
    if not games:
        return 0.0
    wins = sum((oc + 1) / 2 * n for oc, n in results.items() if oc is not None)
    return wins / games


class _ResultTable:
    """Per-agent outcome histograms, total and per seat pattern."""

    def __init__(self, num_agents):
        self.total = [{} for _ in range(num_agents)]
        self.by_pattern = [{} for _ in range(num_agents)]

    def ensure_pattern(self, pattern):
        for per_agent in self.by_pattern:
            per_agent.setdefault(pattern, {})

    def add(self, players, agent_ids, pattern, outcome):
        for seat, p in enumerate(players):
            aid, oc = agent_ids[seat], outcome[p]
            bucket = self.by_pattern[aid][pattern]
            bucket[oc] = bucket.get(oc, 0) + 1
            self.total[aid][oc] = self.total[aid].get(oc, 0) + 1

    def report(self):
        for aid, per_pattern in enumerate(self.by_pattern):
            print('---agent %d---' % aid)
            for pattern, hist in per_pattern.items():
                ordered = {k: hist[k] for k in sorted(hist, reverse=True)}
                print(pattern, ordered, wp_func(hist))
            total = self.total[aid]
            print('total', {k: total[k] for k in sorted(total, reverse=True)},
                  wp_func(total))


def _seating_plan(num_agents, num_games, pat_idx):
    """Seat assignments for one pattern: two-player games split first/second
    evenly (suffix -F/-S); larger games shuffle seats per game."""
    plans = []
    for i in range(num_games):
        if num_agents == 2:
            if i < (num_games + 1) // 2:
                plans.append((pat_idx + '-F', [0, 1]))
            else:
                plans.append((pat_idx + '-S', [1, 0]))
        else:
            plans.append((pat_idx,
                          random.sample(range(num_agents), num_agents)))
    return plans


def eval_process_mp_child(agents, critic, env_args, index, in_queue,
                          out_queue, seed, show=False):
    """One evaluation worker: plays queued games until the poison pill."""
    random.seed(seed + index)
    env = make_env({**env_args, 'id': index})
    while True:
        job = in_queue.get()
        if job is None:
            break
        game_no, agent_ids, pattern, game_args = job
        print('*** Game %d ***' % game_no)
        seat_map = {env.players()[seat]: agents[aid]
                    for seat, aid in enumerate(agent_ids)}
        runner = exec_network_match \
            if isinstance(next(iter(seat_map.values())), NetworkAgent) \
            else exec_match
        results = runner(env, seat_map, critic, show=show,
                         game_args=game_args)
        out_queue.put((pattern, agent_ids, results))
    out_queue.put(None)


def evaluate_mp(env, agents, critic, env_args, args_patterns, num_process,
                num_games, seed):
    """Offline evaluation: enqueue every (pattern x game) job, fan out over
    worker processes, aggregate outcome histograms."""
    in_queue, out_queue = mp.Queue(), mp.Queue()
    table = _ResultTable(len(agents))
    print('total games = %d' % (len(args_patterns) * num_games))
    time.sleep(0.1)

    game_no = 0
    for pat_idx, game_args in args_patterns.items():
        for pattern, agent_ids in _seating_plan(len(agents), num_games,
                                                pat_idx):
            table.ensure_pattern(pattern)
            in_queue.put((game_no, agent_ids, pattern, game_args))
            game_no += 1

    network_mode = agents[0] is None
    if network_mode:
        per_process = network_match_acception(
            num_process, env_args, len(agents), NETWORK_MATCH_PORT)
    else:
        per_process = [agents] * num_process

    for i in range(num_process):
        in_queue.put(None)
        child_args = (per_process[i], critic, env_args, i, in_queue,
                      out_queue, seed)
        if num_process > 1:
            mp.Process(target=eval_process_mp_child, args=child_args).start()
            if network_mode:
                for agent in per_process[i]:
                    agent.conn.close()
        else:
            eval_process_mp_child(*child_args, show=True)

    finished = 0
    while finished < num_process:
        item = out_queue.get()
        if item is None:
            finished += 1
            continue
        pattern, agent_ids, results = item
        outcome = results.get('result') if results else None
        if outcome is not None:
            table.add(env.players(), agent_ids, pattern, outcome)

    table.report()


def network_match_acception(n, env_args, num_agents, port):
    """Accept n*num_agents client connections, grouping each consecutive
    ``num_agents`` into one match; the first client of each group receives
    the env spec."""
    groups, pending = [], []
    for conn in accept_socket_connections(port):
        pending.append(conn)
        if len(pending) == num_agents:
            pending[0].send(env_args)
            groups.append([NetworkAgent(c) for c in pending])
            pending = []
            if len(groups) == n:
                break
    return groups


# -- ONNX inference ----------------------------------------------------------

class OnnxModel:
    """onnxruntime single-thread session with the hidden* name convention
    for recurrent state (scripts/make_onnx_model.py export format)."""

    def __init__(self, model_path):
        self.model_path = model_path
        self.session = None

    def _session(self):
        if self.session is None:
            import os
            os.environ['OMP_NUM_THREADS'] = '1'
            os.environ['OMP_WAIT_POLICY'] = 'PASSIVE'
            import onnxruntime
            opts = onnxruntime.SessionOptions()
            opts.intra_op_num_threads = 1
            opts.inter_op_num_threads = 1
            opts.execution_mode = onnxruntime.ExecutionMode.ORT_SEQUENTIAL
            self.session = onnxruntime.InferenceSession(
                self.model_path, sess_options=opts)
        return self.session

    def init_hidden(self, batch_size=None):
        import numpy as np
        dtypes = {'tensor(float)': np.float32, 'tensor(int64)': np.int64}
        shape_head = list(batch_size) if batch_size is not None else []
        hidden = [np.zeros(shape_head + list(spec.shape[1:]),
                           dtype=dtypes[spec.type])
                  for spec in self._session().get_inputs()
                  if spec.name.startswith('hidden')]
        return hidden or None

    def inference(self, x, hidden=None, batch_input=False):
        import numpy as np
        from .util import map_r
        session = self._session()
        names = [spec.name for spec in session.get_inputs()]
        feeds = {}

        def bind(leaf):
            feeds[names[len(feeds)]] = leaf if batch_input \
                else np.expand_dims(leaf, 0)

        map_r(x, bind)
        if hidden is not None:
            map_r(hidden, bind)
        raw = session.run(None, feeds)
        if not batch_input:
            raw = [o.squeeze(0) for o in raw]
        outputs = {spec.name: raw[i]
                   for i, spec in enumerate(session.get_outputs())}
        hidden_out = [outputs.pop(k) for k in list(outputs)
                      if k.startswith('hidden')]
        return {**outputs, 'hidden': hidden_out or None}


def load_model(model_path, model=None):
    if model_path.endswith('.onnx'):
        return OnnxModel(model_path)
    assert model is not None
    import torch
    from .model import ModelWrapper
    sd = torch.load(model_path)
    if hasattr(model, 'load_reference_state_dict'):
        # nets whose saved layout is the REFERENCE layout (e.g. GeeseNet
        # exports explicit under-BN conv biases) load through their
        # fold-aware loader; it is exact for both layouts
        model.load_reference_state_dict(sd)
    else:
        model.load_state_dict(sd)
    model.eval()
    return ModelWrapper(model)


# -- CLI modes ---------------------------------------------------------------

def _resolve_agent(spec, env):
    agent = build_agent(spec, env)
    return agent if agent is not None else Agent(load_model(spec, env.net()))


def client_mp_child(env_args, model_path, conn):
    env = make_env(env_args)
    agent = _resolve_agent(model_path, env)
    NetworkAgentClient(agent, env, conn).run()


def eval_main(args, argv):
    env_args = args['env_args']
    prepare_env(env_args)
    env = make_env(env_args)

    model_paths = argv[0].split(':') if len(argv) >= 1 else ['models/latest.pth']
    num_games = int(argv[1]) if len(argv) >= 2 else 100
    num_process = int(argv[2]) if len(argv) >= 3 else 1

    main_agent = _resolve_agent(model_paths[0], env)
    critic = None
    print('%d process, %d games' % (num_process, num_games))
    seed = random.randrange(int(1e8))
    print('seed = %d' % seed)
    opponent = model_paths[1] if len(model_paths) > 1 else 'random'
    agents = [main_agent] + [_resolve_agent(opponent, env)
                             for _ in range(len(env.players()) - 1)]
    evaluate_mp(env, agents, critic, env_args, {'default': {}},
                num_process, num_games, seed)


def eval_server_main(args, argv):
    print('network match server mode')
    env_args = args['env_args']
    prepare_env(env_args)
    env = make_env(env_args)
    num_games = int(argv[0]) if len(argv) >= 1 else 100
    num_process = int(argv[1]) if len(argv) >= 2 else 1
    print('%d process, %d games' % (num_process, num_games))
    seed = random.randrange(int(1e8))
    print('seed = %d' % seed)
    evaluate_mp(env, [None] * len(env.players()), None, env_args,
                {'default': {}}, num_process, num_games, seed)


def eval_client_main(args, argv):
    print('network match client mode')
    while True:
        try:
            host = argv[1] if len(argv) >= 2 else 'localhost'
            conn = connect_socket_connection(host, NETWORK_MATCH_PORT)
            env_args = conn.recv()
        except ConnectionResetError:
            break
        model_path = argv[0] if len(argv) >= 1 else 'models/latest.pth'
        mp.Process(target=client_mp_child,
                   args=(env_args, model_path, conn)).start()
        conn.close()
