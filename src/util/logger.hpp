// Phase timer + 20-bin progress bar on stderr.
// Behavioral parity: reference src/logger.{hpp,cpp} (interval log, bar
// redraw with '\r', cumulative total at teardown).
#pragma once

#include <chrono>
#include <cstdio>
#include <string>

namespace rga {

class Logger {
 public:
  Logger() : time_(0.0), bar_(0), time_point_() {}

  // Starts (or restarts) an interval.
  void log() { time_point_ = std::chrono::steady_clock::now(); }

  // Ends the interval and prints elapsed seconds.
  void log(const std::string& msg) {
    auto elapsed = std::chrono::duration_cast<std::chrono::duration<double>>(
                       std::chrono::steady_clock::now() - time_point_)
                       .count();
    time_ += elapsed;
    fprintf(stderr, "%s %.5lf s\n", msg.c_str(), elapsed);
  }

  // Advances the 20-bin progress bar; prints elapsed on the last bin.
  void bar(const std::string& msg) {
    ++bar_;
    std::string progress(bar_, '=');
    progress.resize(20, ' ');
    fprintf(stderr, "%s [%s] %.5lf s", msg.c_str(), progress.c_str(),
            std::chrono::duration_cast<std::chrono::duration<double>>(
                std::chrono::steady_clock::now() - time_point_)
                .count());
    if (bar_ == 20) {
      auto elapsed = std::chrono::duration_cast<std::chrono::duration<double>>(
                         std::chrono::steady_clock::now() - time_point_)
                         .count();
      time_ += elapsed;
      bar_ = 0;
      fprintf(stderr, "\n");
    } else {
      fprintf(stderr, "\r");
    }
  }

  void total(const std::string& msg) const {
    auto elapsed = std::chrono::duration_cast<std::chrono::duration<double>>(
                       std::chrono::steady_clock::now() - time_point_)
                       .count();
    fprintf(stderr, "%s %.5lf s\n", msg.c_str(), time_ + (bar_ != 0 ? elapsed : 0.0));
  }

 private:
  double time_;
  uint32_t bar_;
  std::chrono::steady_clock::time_point time_point_;
};

}  // namespace rga
