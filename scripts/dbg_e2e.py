import faulthandler, sys, os, tempfile
faulthandler.dump_traceback_later(90, exit=True)
sys.path.insert(0, "/root/repo")
from torchbeast_amd import polybeast_learner
flags = polybeast_learner.parser.parse_args([])
flags.env = "synthetic:4x84x84:6"
flags.savedir = tempfile.mkdtemp()
flags.xpid = "gpue2e"
flags.num_actors = 16
flags.batch_size = 8
flags.unroll_length = 20
flags.total_steps = 8 * 20 * 6
flags.num_learner_threads = 1
flags.num_inference_threads = 1
polybeast_learner.train(flags)
print("TRAIN DONE")
