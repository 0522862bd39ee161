"""cProfile wrapper. Parity: reference ding/utils/profiler_helper.py:12."""
import atexit


def register_profiler(write_profile, pr, folder_path):
    """Flush the profile at interpreter exit (reference profiler_helper.py:8)."""
    atexit.register(write_profile, pr, folder_path)
import cProfile
import os
import pstats


class Profiler:

    def __init__(self):
        self.pr = cProfile.Profile()

    def mkdir(self, directory: str):
        os.makedirs(directory, exist_ok=True)

    def write_profile(self, pr: cProfile.Profile, folder_path: str):
        pr.disable()
        for sort_key, fname in (("tottime", "profile_tottime.txt"), ("cumtime", "profile_cumtime.txt")):
            with open(os.path.join(folder_path, fname), "w") as f:
                ps = pstats.Stats(pr, stream=f).sort_stats(sort_key)
                ps.print_stats()
        pr.dump_stats(os.path.join(folder_path, "profile.prof"))

    def profile(self, folder_path: str = "./tmp") -> None:
        self.mkdir(folder_path)
        self.pr.enable()
        atexit.register(self.write_profile, self.pr, folder_path)
