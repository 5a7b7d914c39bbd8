// conc_main.cpp — hpk_conc: single-GPU stream-concurrency benchmark CLI.
//
// MI355X-native re-design of the reference concurrency driver
// (reference concurency/main.cpp:115-322): same CLI surface
// (mode, repeated --commands lists, --tripcount_C/--globalsize_*/--queues/
// --repetitions/--min_bandwidth/--enable_profiling/--verbose), same
// autotuning idea (linear rescale so all commands take equal time), same
// pass/fail criteria (speedup within 30% of theoretical; optional bandwidth
// floor) and the same "## mode | commands | SUCCESS/FAILURE" log grammar that
// scripts/parse.py consumes. The engine underneath is hipStreams/hipGraphs
// (hpc_patterns_amd/native/conc.hip), not SYCL queues.
//
// Extras over the reference: --copy_kernel (shader-blit copies instead of
// SDMA hipMemcpyAsync), --csv FILE (machine-readable results), and the
// graph/out_of_order mode pair.

#include "../hpc_patterns_amd/native/include/hpk.h"

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <fstream>
#include <iostream>
#include <limits>
#include <map>
#include <set>
#include <sstream>
#include <string>
#include <vector>

namespace {

constexpr double kTolSpeedup = 0.3; // reference main.cpp:12

void help_and_exit(const std::string& bin, const std::string& msg) {
  if (!msg.empty()) std::cout << "ERROR: " << msg << std::endl;
  std::cout
      << "Usage: " << bin << " " << hpk::allowed_modes << "\n"
      << "                [--enable_profiling] [--verbose] [--copy_kernel]\n"
      << "                [--copy_engine auto|shader|sdma] [--mfma]\n"
      << "                [--tripcount_C <tripcount>]\n"
      << "                [--globalsize_{C,A2B} <global_size>]\n"
      << "                [--globalsize_default_memory <floats>]\n"
      << "                [--queues <n_queues>]\n"
      << "                [--repetitions <n_repetitions>]\n"
      << "                [--min_bandwidth <GB/s>]\n"
      << "                [--csv <file>]\n"
      << "                [--commands COMMANDS..]\n"
      << "\n"
      << "COMMAND: C (compute: 64*tripcount FMA per work-item) or A2B memcopy\n"
      << "         with A,B in {M: malloc, D: hipMalloc, H: hipHostMalloc,\n"
      << "         S: hipMallocManaged}. '-1' parameters are auto-tuned so\n"
      << "         every command takes similar time.\n";
  std::exit(1);
}

std::string sanitize(const std::string& cmd) {
  std::string out;
  for (char c : cmd)
    if (c != '2') out += c;
  return out;
}

bool command_ok(const std::string& sc) {
  if (sc == "C") return true;
  if (sc.size() != 2) return false;
  const std::string letters = "MDHS";
  for (char c : sc)
    if (letters.find(c) == std::string::npos) return false;
  // host->host pairs measure nothing on-GPU
  std::set<std::string> banned = {"HM", "MH", "MM", "HH"};
  return banned.find(sc) == banned.end();
}

std::string tuned_param_name(const std::string& cmd) {
  return cmd == "C" ? "tripcount_C" : "globalsize_" + cmd;
}

size_t default_param(const std::string& name, long default_memory) {
  if (name.rfind("globalsize_C", 0) == 0) return 1;
  if (name.rfind("tripcount_C", 0) == 0) return 40000;
  if (name.rfind("globalsize_", 0) == 0) {
    if (default_memory > 0) return (size_t)default_memory;
    return (size_t)(1e9 / sizeof(float)); // ~1 GB of floats
  }
  return 0;
}

std::string time_info(const std::vector<std::string>& cmds, long time_us,
                      const std::map<std::string, size_t>& params,
                      double min_bandwidth, int* bw_errno) {
  size_t bytes = 0;
  for (const auto& c : cmds)
    if (c != "C") bytes += params.at("globalsize_" + c) * sizeof(float);
  std::ostringstream out;
  out << time_us << "us";
  if (bytes) {
    double gbps = 1e-3 * (double)bytes / (double)time_us;
    out << " (" << gbps << " GBytes/s)";
    if (bw_errno) *bw_errno = (min_bandwidth >= 0 && gbps < min_bandwidth) ? -1 : 0;
  }
  return out.str();
}

} // namespace

int main(int argc, char* argv[]) {
  std::vector<std::string> args(argv + 1, argv + argc);
  if (args.empty() || args[0] == "--help" || args[0] == "-h")
    help_and_exit(argv[0], "");

  std::string mode = args[0];
  if (!hpk::mode_is_allowed(mode))
    help_and_exit(argv[0], "unknown mode '" + mode + "'");

  std::map<std::string, long> params_cli = {{"globalsize_C", -1},
                                            {"tripcount_C", -1},
                                            {"globalsize_default_memory", -1}};
  bool enable_profiling = false, verbose = false;
  int copy_engine = hpk::kCopyEngineAuto;
  int n_queues = -1, n_repetitions = 10;
  double min_bandwidth = -1;
  std::string csv_path;

  std::vector<std::vector<std::string>> l_commands;
  std::vector<std::string> commands;
  args.push_back("--commands"); // flush trailing list

  for (size_t i = 1; i < args.size(); ++i) {
    const std::string& s = args[i];
    auto next = [&](const char* opt) -> std::string {
      if (++i >= args.size())
        help_and_exit(argv[0], std::string("missing value for ") + opt);
      return args[i];
    };
    if (s == "--enable_profiling") enable_profiling = true;
    else if (s == "--mfma") params_cli["payload_C_mfma"] = 1;
    else if (s == "--verbose") verbose = true;
    else if (s == "--copy_kernel") copy_engine = hpk::kCopyEngineShader;
    else if (s == "--copy_engine") {
      std::string v = next("--copy_engine");
      if (v == "auto") copy_engine = hpk::kCopyEngineAuto;
      else if (v == "shader") copy_engine = hpk::kCopyEngineShader;
      else if (v == "sdma") copy_engine = hpk::kCopyEngineSdma;
      else help_and_exit(argv[0], "copy_engine must be auto|shader|sdma");
    }
    else if (s == "--queues") n_queues = std::stoi(next("--queues"));
    else if (s == "--repetitions") n_repetitions = std::stoi(next("--repetitions"));
    else if (s == "--min_bandwidth") min_bandwidth = std::stod(next("--min_bandwidth"));
    else if (s == "--csv") csv_path = next("--csv");
    else if (s.rfind("--tripcount_", 0) == 0 || s.rfind("--globalsize_", 0) == 0)
      params_cli[s.substr(2)] = std::stol(next(s.c_str()));
    else if (s == "--commands") {
      if (!commands.empty()) {
        l_commands.push_back(commands);
        commands.clear();
      }
    } else if (s.rfind("-", 0) == 0)
      help_and_exit(argv[0], "unsupported option '" + s + "'");
    else {
      std::string sc = sanitize(s);
      if (!command_ok(sc))
        help_and_exit(argv[0], "unsupported COMMAND '" + s + "'");
      commands.push_back(sc);
    }
  }
  if (l_commands.empty())
    help_and_exit(argv[0], "need --commands (C, M2D, D2M, H2D, D2H, D2D, ...)");

  // Collect every parameter mentioned by any list; resolve defaults.
  for (const auto& lc : l_commands)
    for (const auto& c : lc)
      if (c != "C") params_cli.try_emplace("globalsize_" + c, -1);

  std::map<std::string, size_t> params;
  long default_memory = params_cli["globalsize_default_memory"];
  for (const auto& [k, v] : params_cli)
    params[k] = (v == -1) ? default_param(k, default_memory) : (size_t)v;

  std::set<std::string> uniq;
  for (const auto& lc : l_commands) uniq.insert(lc.begin(), lc.end());

  // ---- autotune: rescale each auto (-1) parameter so all commands take the
  // time of the fastest copy command (linear model; same strategy as the
  // reference main.cpp:226-258, re-implemented) ----
  bool need_tune = false;
  for (const auto& c : uniq)
    need_tune |= (params_cli[tuned_param_name(c)] == -1);

  if (need_tune && uniq.size() > 1) {
    std::cout << "# Performing Autotuning to Balance Commands Times" << std::endl;
    std::vector<std::string> uniq_vec(uniq.begin(), uniq.end());
    auto base = hpk::conc_bench("serial", uniq_vec, params, false, n_queues,
                                n_repetitions, verbose, copy_engine);
    long target = std::numeric_limits<long>::max();
    for (size_t i = 0; i < uniq_vec.size(); ++i)
      if (uniq_vec[i] != "C")
        target = std::min(target, base.per_cmd_us[i]);
    if (target == std::numeric_limits<long>::max())
      target = *std::max_element(base.per_cmd_us.begin(), base.per_cmd_us.end());
    for (size_t i = 0; i < uniq_vec.size(); ++i) {
      const std::string pname = tuned_param_name(uniq_vec[i]);
      if (params_cli[pname] == -1 && base.per_cmd_us[i] > 0) {
        size_t cur = params[pname];
        size_t scaled =
            (size_t)((double)target / (double)base.per_cmd_us[i] * (double)cur);
        params[pname] = std::max<size_t>(scaled, 1);
      }
    }
  }

  std::cout << "Parameters used:" << std::endl;
  for (const auto& c : uniq) {
    const std::string p = tuned_param_name(c);
    std::cout << "  " << p << ": " << params[p] << std::endl;
    if (c == "C")
      std::cout << "  globalsize_C: " << params["globalsize_C"] << std::endl;
  }

  std::ofstream csv;
  if (!csv_path.empty()) {
    csv.open(csv_path);
    csv << "mode,commands,serial_us,concurrent_us,theoretical_speedup,"
           "speedup,bandwidth_gbps,verdict\n";
  }

  int exit_code = 0;
  for (const auto& cmds : l_commands) {
    std::ostringstream label;
    label << mode << " | ";
    for (const auto& c : cmds) label << c << " ";
    std::cout << "# " << label.str() << "| Starting Benchmarking..." << std::endl;

    auto serial = hpk::conc_bench("serial", cmds, params, enable_profiling,
                                  n_queues, n_repetitions, verbose,
                                  copy_engine);
    std::cout << "Minimum Measured Total Time Serial: " << serial.total_us
              << "us" << std::endl;
    for (size_t i = 0; i < cmds.size(); ++i) {
      std::cout << "  Minimum Time Command " << i << " (" << cmds[i] << "): "
                << time_info({cmds[i]}, serial.per_cmd_us[i], params, -1,
                             nullptr)
                << std::endl;
      if (enable_profiling && serial.per_cmd_dev_ms[i] >= 0.0)
        std::cout << "    Device Time (hipEvent): " << serial.per_cmd_dev_ms[i]
                  << "ms" << std::endl;
    }
    long max_cmd =
        *std::max_element(serial.per_cmd_us.begin(), serial.per_cmd_us.end());
    double max_speedup = (double)serial.total_us / (double)std::max(max_cmd, 1L);
    std::cout << "Maximum Theoretical Speedup: " << max_speedup << "x"
              << std::endl;
    if (cmds.size() > 1 && max_speedup <= 1.50)
      std::cerr << "  WARNING: Large Unbalance Between Commands" << std::endl;

    auto conc = hpk::conc_bench(mode, cmds, params, enable_profiling, n_queues,
                                n_repetitions, verbose, copy_engine);
    int bw_errno = 0;
    std::string conc_info =
        time_info(cmds, conc.total_us, params, min_bandwidth, &bw_errno);
    std::cout << "Minimum Measured Total Time //: " << conc_info << std::endl;
    if (enable_profiling) {
      // per-command device times in the CONCURRENT run (graph modes report
      // these via event-record graph nodes — r1 left them unmeasured)
      for (size_t i = 0; i < cmds.size(); ++i)
        if (conc.per_cmd_dev_ms[i] >= 0.0)
          std::cout << "  // Device Time Command " << i << " (" << cmds[i]
                    << "): " << conc.per_cmd_dev_ms[i] << "ms" << std::endl;
    }
    double speedup = (double)serial.total_us / (double)std::max(conc.total_us, 1L);
    std::cout << "Speedup Relative to Serial: " << speedup << "x" << std::endl;

    std::string verdict;
    if (bw_errno != 0) {
      verdict = "FAILURE: Minimum Bandwidth not reached";
      exit_code = 1;
    } else if (max_speedup >= (1.0 + kTolSpeedup) * speedup) {
      verdict = "FAILURE: Far from Theoretical Speedup";
      exit_code = 1;
    } else {
      verdict = "SUCCESS: Close from Theoretical Speedup";
    }
    std::cout << "## " << label.str() << "| " << verdict << std::endl;

    if (csv.is_open()) {
      size_t bytes = 0;
      for (const auto& c : cmds)
        if (c != "C") bytes += params["globalsize_" + c] * sizeof(float);
      double gbps =
          bytes ? 1e-3 * (double)bytes / (double)conc.total_us : 0.0;
      std::string cmd_join;
      for (const auto& c : cmds) cmd_join += (cmd_join.empty() ? "" : " ") + c;
      csv << mode << "," << cmd_join << "," << serial.total_us << ","
          << conc.total_us << "," << max_speedup << "," << speedup << ","
          << gbps << "," << verdict.substr(0, verdict.find(':')) << "\n";
    }
  }
  return exit_code;
}
