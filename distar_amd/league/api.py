"""League HTTP API (reference `ctools/worker/league/league_api.py:14-60`):
register_learner / learner_send_train_info / actor_ask_for_job /
actor_send_result + debug endpoints, served by the stdlib JSON HTTP server."""
from ..utils.http import JsonHttpServer, pick_unused_port


def create_league_server(league, host='0.0.0.0', port=None):
    port = port or pick_unused_port()

    def show_payoff(body):
        pid = body.get('player_id')
        players = ([league.all_players[pid]] if pid
                   else list(league.active_players.values()))
        return {p.player_id: p.payoff.stat_info_dict for p in players}

    def add_hist_player(body):
        from .player import HistoricalPlayer
        league.set_hist_player(HistoricalPlayer(
            checkpoint_path=body['checkpoint_path'],
            player_id=body['player_id'],
            pipeline=body.get('pipeline', 'default'),
            frac_id=body.get('frac_id', 1),
            z_path=body.get('z_path', '3map.json'),
            z_prob=body.get('z_prob', 0.)))
        return {'ok': True}

    def show_elo(body):
        return {'text': league.elo.elo_text(),
                'ratings': dict(league.elo.ratings)}

    def snapshot_player(body):
        player = league.active_players[body['player_id']]
        player.snapshot_flag = True
        return {'ok': True}

    def reset_player(body):
        player = league.active_players[body['player_id']]
        player.reset_flag = True
        return {'ok': True}

    def _players_of(body):
        pid = body.get('player_id')
        if pid:
            return [league.all_players[pid]]
        return list(league.active_players.values())

    def show_dist_stat(body):
        return {p.player_id: p.dist_stat.stat_info_dict
                for p in _players_of(body) if hasattr(p, 'dist_stat')}

    def show_cum_stat(body):
        return {p.player_id: p.cum_stat.stat_info_dict
                for p in _players_of(body) if hasattr(p, 'cum_stat')}

    def show_unit_num_stat(body):
        return {p.player_id: p.unit_num_stat.stat_info_dict
                for p in _players_of(body) if hasattr(p, 'unit_num_stat')}

    def show_trueskill(body):
        return {'mu': dict(league.trueskill.mu),
                'sigma': dict(league.trueskill.sigma)}

    def show_config(body):
        def plain(x):
            if isinstance(x, dict):
                return {k: plain(v) for k, v in x.items()}
            if isinstance(x, (list, tuple)):
                return [plain(v) for v in x]
            return x if isinstance(x, (int, float, bool, str, type(None))) \
                else str(x)
        return plain(dict(league.cfg))

    def update_config(body):
        from ..utils.config import Config, deep_merge_dicts
        league.cfg = deep_merge_dicts(league.cfg,
                                      Config(body.get('overrides', {})))
        return {'ok': True}

    def remove_hist_player(body):
        removed = league.historical_players.pop(body['player_id'], None)
        return {'ok': removed is not None}

    def display_player(body):
        out = {}
        for p in _players_of(body):
            out[p.player_id] = {
                'checkpoint_path': p.checkpoint_path,
                'pipeline': p.pipeline, 'frac_id': p.frac_id,
                'z_path': p.z_path, 'z_prob': p.z_prob,
                'total_agent_step': getattr(p, 'total_agent_step', 0),
                'total_game_count': getattr(p, 'total_game_count', 0),
            }
        return out

    server = JsonHttpServer({
        '/league/register_learner': league.deal_with_register_learner,
        '/league/learner_send_train_info': league.deal_with_learner_send_train_info,
        '/league/actor_ask_for_job': league.deal_with_actor_ask_for_job,
        '/league/actor_send_result':
            lambda body: {'ok': league.deal_with_actor_send_result(body)},
        '/league/show_payoff': show_payoff,
        '/league/show_elo': show_elo,
        '/league/add_hist_player': add_hist_player,
        '/league/snapshot_player': snapshot_player,
        '/league/reset_player': reset_player,
        '/league/save_resume': lambda body: {'path': league.save_resume()},
        '/league/show_dist_stat': show_dist_stat,
        '/league/show_cum_stat': show_cum_stat,
        '/league/show_unit_num_stat': show_unit_num_stat,
        '/league/show_trueskill': show_trueskill,
        '/league/show_config': show_config,
        '/league/update_config': update_config,
        '/league/remove_hist_player': remove_hist_player,
        '/league/display_player': display_player,
    }, host=host, port=port)
    return server
