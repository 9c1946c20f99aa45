#!/usr/bin/env bash
# Tiny-scale smoke run of every system entry point (parity role of the
# reference's bash_scripts/run-algorithms.sh). Exits non-zero on the first
# failing system.
set -euo pipefail
TINY="arch.total_num_envs=8 arch.total_timesteps=null arch.num_updates=2 \
arch.num_evaluation=1 arch.num_eval_episodes=2 system.rollout_length=8 \
logger.loggers=[] logger.checkpointing.save_model=false"

run() {
  echo "=== $1 $2"
  python -m "$1" $TINY ${@:3}
}

run stoix_amd.systems.ppo.ff_ppo              "" env=classic/cartpole system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.ff_ppo              "" env=classic/pendulum network=mlp_continuous system=ppo/ff_ppo_continuous system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.ff_ppo_penalty      "" system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.ff_dpo              "" system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.rec_ppo             "" system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.vpg.ff_reinforce        ""
run stoix_amd.systems.q_learning.ff_dqn       "" system.batch_size=8 system.buffer_size=512 system.warmup_steps=8
run stoix_amd.systems.q_learning.ff_ddqn      "" system.batch_size=8 system.buffer_size=512 system.warmup_steps=8
run stoix_amd.systems.q_learning.ff_mdqn      "" system.batch_size=8 system.buffer_size=512 system.warmup_steps=8
run stoix_amd.systems.q_learning.ff_dqn_reg   "" system.batch_size=8 system.buffer_size=512 system.warmup_steps=8
run stoix_amd.systems.q_learning.ff_c51       "" system.batch_size=8 system.buffer_size=512 system.warmup_steps=8
run stoix_amd.systems.q_learning.ff_qr_dqn    "" system.batch_size=8 system.buffer_size=512 system.warmup_steps=8
run stoix_amd.systems.q_learning.ff_pqn       "" system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.q_learning.ff_rainbow   "" system.batch_size=16 system.buffer_size=512 system.warmup_steps=16 system.n_step=3 system.epochs=2
run stoix_amd.systems.q_learning.rec_r2d2     "" system.batch_size=8 system.buffer_size=512 system.sample_sequence_length=8 system.burn_in_length=2 system.n_step=2 system.rollout_length=10 system.epochs=1
run stoix_amd.systems.sac.ff_sac              "" system.batch_size=8 system.buffer_size=512 system.warmup_steps=8
run stoix_amd.systems.ddpg.ff_ddpg            "" system.batch_size=8 system.buffer_size=512 system.warmup_steps=8
run stoix_amd.systems.ddpg.ff_td3             "" system.batch_size=8 system.buffer_size=512 system.warmup_steps=8
run stoix_amd.systems.ddpg.ff_d4pg            "" system.batch_size=8 system.buffer_size=512 system.warmup_steps=8
run stoix_amd.systems.mpo.ff_mpo              "" system.epochs=1 system.batch_size=8 system.buffer_size=256 system.sample_sequence_length=4 system.num_samples=4
run stoix_amd.systems.mpo.ff_vmpo             ""
run stoix_amd.systems.awr.ff_awr              "" system.epochs=1 system.batch_size=8 system.buffer_size=256 system.sample_sequence_length=4
run stoix_amd.systems.search.ff_az            "" system.num_simulations=6 system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.search.ff_sampled_az    "" system.num_simulations=4 system.num_sampled_actions=4 system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.search.ff_mz            "" system.num_simulations=4 system.epochs=1 system.unroll_steps=2 system.n_step=2 system.batch_size=8 system.buffer_size=256
run stoix_amd.systems.search.ff_sampled_mz    "" system.num_simulations=4 system.num_sampled_actions=4 system.epochs=1 system.unroll_steps=2 system.n_step=2 system.batch_size=8 system.buffer_size=256
run stoix_amd.systems.spo.ff_spo              "" system.num_particles=4 system.search_depth=2 system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.disco_rl.ff_disco103    "" system.num_minibatches=2 system.epochs=1 system.disco_rule.num_bins=21 system.disco_rule.net.prediction_size=16
run stoix_amd.systems.ppo.ff_ppo              "xland" env=xland_minigrid/goal_grid system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.ff_ppo              "craftax" env=craftax/crafting system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.rec_ppo             "popjym" env=popjym/stateless_cartpole system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.rec_ppo             "popjym-mem" env=popjym/repeat_first_easy system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.rec_ppo             "arcade" env=popgym_arcade/noisy_cartpole system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.ff_ppo              "doorkey" env=navix/doorkey system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.ff_ppo              "swingup" env=mjc_playground/cartpole_swingup system=ppo/ff_ppo_continuous network=mlp_continuous system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.ff_ppo              "reacher" env=kinetix/reacher system=ppo/ff_ppo_continuous network=mlp_continuous system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.ff_ppo              "vizdoom" env=envpool/vizdoom_basic network=cnn system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.ppo.ff_ppo              "battlezone" env=envpool/battlezone network=cnn system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.impala.sebulba_ff_impala "" arch.actor.actor_per_device=2 system.num_minibatches=2
run stoix_amd.systems.ppo.sebulba_ff_ppo      "" arch.actor.actor_per_device=2 system.num_minibatches=2 system.epochs=1
run stoix_amd.systems.impala.sebulba_ff_impala_shared_torso "" arch.actor.actor_per_device=2 system.num_minibatches=2
echo "ALL SYSTEMS OK"
