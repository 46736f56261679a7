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
    }, host=host, port=port)
    return server
