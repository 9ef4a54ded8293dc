// Native config: srtb_config.cfg-compatible parsing with arithmetic
// expression values (reference program_options.hpp + exprgrammar; the Python
// twin is srtb_amd/config.py — keep key sets in sync).
#pragma once

#include <cctype>
#include <cmath>
#include <cstdint>
#include <fstream>
#include <functional>
#include <map>
#include <sstream>
#include <stdexcept>
#include <string>
#include <vector>

namespace srtb_app {

// ---- arithmetic expression evaluator (grammar of srtb_amd/utils/expr.py:
// + - * / % ** parens, unary +/- binding tighter than **, case-insensitive
// functions/constants) ----
class Expr {
 public:
  static double eval(const std::string& text) {
    Expr e(text);
    double v = e.expr();
    e.skip_ws();
    if (e.pos_ != e.t_.size())
      throw std::runtime_error("expr: trailing characters in '" + text + "'");
    return v;
  }

  static long long eval_int(const std::string& text) {
    const double v = eval(text);
    const double r = std::round(v);
    if (std::fabs(v - r) > 1e-6 * std::max(1.0, std::fabs(v)))
      throw std::runtime_error("expr: expected integer from '" + text + "'");
    return (long long)r;
  }

 private:
  explicit Expr(const std::string& t) : t_(t) {}

  void skip_ws() {
    while (pos_ < t_.size() && (t_[pos_] == ' ' || t_[pos_] == '\t')) ++pos_;
  }
  char peek() {
    skip_ws();
    return pos_ < t_.size() ? t_[pos_] : '\0';
  }
  void expect(char c) {
    if (peek() != c)
      throw std::runtime_error(std::string("expr: expected '") + c + "'");
    ++pos_;
  }

  double expr() {
    double v = term();
    while (true) {
      const char c = peek();
      if (c == '+') { ++pos_; v += term(); }
      else if (c == '-') { ++pos_; v -= term(); }
      else return v;
    }
  }
  double term() {
    double v = power();
    while (true) {
      const char c = peek();
      if (c == '*') {
        if (pos_ + 1 < t_.size() && t_[pos_ + 1] == '*') return v;
        ++pos_; v *= power();
      } else if (c == '/') { ++pos_; v /= power(); }
      else if (c == '%') { ++pos_; v = std::fmod(v, power()); }
      else return v;
    }
  }
  double power() {
    double v = unary();
    skip_ws();
    if (pos_ + 1 < t_.size() && t_[pos_] == '*' && t_[pos_ + 1] == '*') {
      pos_ += 2;
      return std::pow(v, power());
    }
    return v;
  }
  double unary() {
    double sign = 1.0;
    while (true) {
      const char c = peek();
      if (c == '-') { sign = -sign; ++pos_; }
      else if (c == '+') ++pos_;
      else break;
    }
    return sign * atom();
  }
  double atom() {
    const char c = peek();
    if (c == '(') {
      ++pos_;
      const double v = expr();
      expect(')');
      return v;
    }
    if (std::isalpha((unsigned char)c) || c == '_') return symbol();
    return number();
  }
  double symbol() {
    skip_ws();
    size_t s = pos_;
    while (pos_ < t_.size() &&
           (std::isalnum((unsigned char)t_[pos_]) || t_[pos_] == '_'))
      ++pos_;
    std::string name = t_.substr(s, pos_ - s);
    for (auto& ch : name) ch = (char)std::tolower((unsigned char)ch);
    if (peek() == '(') {
      ++pos_;
      const double a = expr();
      static const std::map<std::string, double (*)(double)> uf = {
          {"abs", std::fabs}, {"acos", std::acos}, {"asin", std::asin},
          {"atan", std::atan}, {"ceil", std::ceil}, {"cos", std::cos},
          {"cosh", std::cosh}, {"exp", std::exp}, {"floor", std::floor},
          {"log", std::log}, {"log10", std::log10}, {"sin", std::sin},
          {"sinh", std::sinh}, {"sqrt", std::sqrt}, {"tan", std::tan},
          {"tanh", std::tanh}};
      auto u = uf.find(name);
      if (u != uf.end()) {
        expect(')');
        return u->second(a);
      }
      expect(',');
      const double b = expr();
      expect(')');
      if (name == "atan2") return std::atan2(a, b);
      if (name == "max") return std::max(a, b);
      if (name == "min") return std::min(a, b);
      if (name == "pow") return std::pow(a, b);
      throw std::runtime_error("expr: unknown function " + name);
    }
    if (name == "pi") return M_PI;
    if (name == "e") return M_E;
    if (name == "epsilon") return 2.220446049250313e-16;
    if (name == "digits") return 53;
    if (name == "digits10") return 15;
    throw std::runtime_error("expr: unknown symbol " + name);
  }
  double number() {
    skip_ws();
    size_t s = pos_;
    while (pos_ < t_.size() && std::isdigit((unsigned char)t_[pos_])) ++pos_;
    if (pos_ < t_.size() && t_[pos_] == '.') {
      ++pos_;
      while (pos_ < t_.size() && std::isdigit((unsigned char)t_[pos_])) ++pos_;
    }
    if (pos_ < t_.size() && (t_[pos_] == 'e' || t_[pos_] == 'E')) {
      size_t j = pos_ + 1;
      if (j < t_.size() && (t_[j] == '+' || t_[j] == '-')) ++j;
      if (j < t_.size() && std::isdigit((unsigned char)t_[j])) {
        pos_ = j;
        while (pos_ < t_.size() && std::isdigit((unsigned char)t_[pos_]))
          ++pos_;
      }
    }
    if (pos_ == s) throw std::runtime_error("expr: expected number");
    return std::stod(t_.substr(s, pos_ - s));
  }

  std::string t_;
  size_t pos_ = 0;
};

// ---- runtime config (defaults = reference config.hpp:80-249) ----
struct Config {
  std::string config_file_name = "srtb_config.cfg";
  size_t baseband_input_count = 1ull << 28;
  int baseband_input_bits = 8;
  std::string baseband_format_type = "simple";
  double baseband_freq_low = 1000.0;
  double baseband_bandwidth = 500.0;
  double baseband_sample_rate = 1e9;
  bool baseband_reserve_sample = true;
  double dm = 0.0;
  std::vector<std::string> udp_receiver_address = {"10.0.1.2"};
  std::vector<int> udp_receiver_port = {12004};
  std::vector<int> udp_receiver_cpu_preferred = {0};
  std::string input_file_path;
  size_t input_file_offset_bytes = 0;
  std::string baseband_output_file_prefix = "srtb_baseband_output_";
  bool baseband_write_all = false;
  std::string fft_fftw_wisdom_path = "srtb_fftw_wisdom.txt";
  double mitigate_rfi_average_method_threshold = 10.0;
  double mitigate_rfi_spectral_kurtosis_threshold = 1.1;
  std::string mitigate_rfi_freq_list;
  size_t spectrum_sum_count = 1;
  size_t spectrum_channel_count = 1ull << 15;
  double signal_detect_signal_noise_threshold = 6.0;
  double signal_detect_channel_threshold = 0.9;
  size_t signal_detect_max_boxcar_length = 1024;
  size_t thread_query_work_wait_time = 1000;
  bool gui_enable = false;
  size_t gui_pixmap_width = 1920;
  size_t gui_pixmap_height = 1080;
  int log_level = 3;

  void assign(const std::string& key, const std::string& raw);
  void parse_file(const std::string& path);
  // cmd > cfg-file > defaults (reference README.md:146)
  void parse_args(int argc, char** argv);
  std::string dump() const;
};

inline std::string trim(const std::string& s) {
  size_t a = s.find_first_not_of(" \t\r\n");
  size_t b = s.find_last_not_of(" \t\r\n");
  return a == std::string::npos ? "" : s.substr(a, b - a + 1);
}

inline std::vector<std::string> split_list(const std::string& s) {
  std::vector<std::string> out;
  std::stringstream ss(s);
  std::string item;
  while (std::getline(ss, item, ',')) {
    item = trim(item);
    if (!item.empty()) out.push_back(item);
  }
  return out;
}

inline void Config::assign(const std::string& key, const std::string& raw0) {
  const std::string raw = trim(raw0);
  auto I = [&] { return (size_t)Expr::eval_int(raw); };
  auto D = [&] { return Expr::eval(raw); };
  auto B = [&] { return Expr::eval_int(raw) != 0; };
  if (key == "config_file_name") config_file_name = raw;
  else if (key == "baseband_input_count") baseband_input_count = I();
  else if (key == "baseband_input_bits") baseband_input_bits = (int)Expr::eval_int(raw);
  else if (key == "baseband_format_type") baseband_format_type = raw;
  else if (key == "baseband_freq_low") baseband_freq_low = D();
  else if (key == "baseband_bandwidth") baseband_bandwidth = D();
  else if (key == "baseband_sample_rate") baseband_sample_rate = D();
  else if (key == "baseband_reserve_sample") baseband_reserve_sample = B();
  else if (key == "dm") dm = D();
  else if (key == "udp_receiver_address") udp_receiver_address = split_list(raw);
  else if (key == "udp_receiver_port") {
    udp_receiver_port.clear();
    for (auto& p : split_list(raw))
      udp_receiver_port.push_back((int)Expr::eval_int(p));
  } else if (key == "udp_receiver_cpu_preferred") {
    udp_receiver_cpu_preferred.clear();
    for (auto& p : split_list(raw))
      udp_receiver_cpu_preferred.push_back((int)Expr::eval_int(p));
  } else if (key == "input_file_path") input_file_path = raw;
  else if (key == "input_file_offset_bytes") input_file_offset_bytes = I();
  else if (key == "baseband_output_file_prefix") baseband_output_file_prefix = raw;
  else if (key == "baseband_write_all") baseband_write_all = B();
  else if (key == "fft_fftw_wisdom_path") fft_fftw_wisdom_path = raw;
  else if (key == "mitigate_rfi_average_method_threshold") mitigate_rfi_average_method_threshold = D();
  else if (key == "mitigate_rfi_spectral_kurtosis_threshold") mitigate_rfi_spectral_kurtosis_threshold = D();
  else if (key == "mitigate_rfi_freq_list") mitigate_rfi_freq_list = raw;
  else if (key == "spectrum_sum_count") spectrum_sum_count = I();
  else if (key == "spectrum_channel_count") spectrum_channel_count = I();
  else if (key == "signal_detect_signal_noise_threshold") signal_detect_signal_noise_threshold = D();
  else if (key == "signal_detect_channel_threshold") signal_detect_channel_threshold = D();
  else if (key == "signal_detect_max_boxcar_length") signal_detect_max_boxcar_length = I();
  else if (key == "thread_query_work_wait_time") thread_query_work_wait_time = I();
  else if (key == "gui_enable") gui_enable = B();
  else if (key == "gui_pixmap_width") gui_pixmap_width = I();
  else if (key == "gui_pixmap_height") gui_pixmap_height = I();
  else if (key == "log_level") log_level = (int)Expr::eval_int(raw);
  else throw std::runtime_error("unknown config key: " + key);
}

inline void Config::parse_file(const std::string& path) {
  std::ifstream f(path);
  if (!f) throw std::runtime_error("cannot open config file " + path);
  std::string line;
  while (std::getline(f, line)) {
    const size_t h = line.find('#');
    if (h != std::string::npos) line = line.substr(0, h);
    line = trim(line);
    if (line.empty()) continue;
    const size_t eq = line.find('=');
    if (eq == std::string::npos)
      throw std::runtime_error("bad config line: " + line);
    assign(trim(line.substr(0, eq)), line.substr(eq + 1));
  }
}

inline void Config::parse_args(int argc, char** argv) {
  std::vector<std::pair<std::string, std::string>> pairs;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    if (a.rfind("--", 0) != 0)
      throw std::runtime_error("unexpected argument " + a);
    a = a.substr(2);
    const size_t eq = a.find('=');
    if (eq != std::string::npos) {
      pairs.emplace_back(a.substr(0, eq), a.substr(eq + 1));
    } else {
      if (i + 1 >= argc) throw std::runtime_error("missing value for " + a);
      pairs.emplace_back(a, argv[++i]);
    }
  }
  for (auto& [k, v] : pairs)
    if (k == "config_file_name") {
      config_file_name = trim(v);
      parse_file(config_file_name);
    }
  for (auto& [k, v] : pairs)
    if (k != "config_file_name") assign(k, v);
}

inline std::string Config::dump() const {
  std::ostringstream o;
  o << "baseband_input_count = " << baseband_input_count << "\n"
    << "baseband_input_bits = " << baseband_input_bits << "\n"
    << "baseband_format_type = " << baseband_format_type << "\n"
    << "baseband_freq_low = " << baseband_freq_low << "\n"
    << "baseband_bandwidth = " << baseband_bandwidth << "\n"
    << "baseband_sample_rate = " << baseband_sample_rate << "\n"
    << "baseband_reserve_sample = " << baseband_reserve_sample << "\n"
    << "dm = " << dm << "\n"
    << "spectrum_channel_count = " << spectrum_channel_count << "\n"
    << "mitigate_rfi_average_method_threshold = "
    << mitigate_rfi_average_method_threshold << "\n"
    << "mitigate_rfi_spectral_kurtosis_threshold = "
    << mitigate_rfi_spectral_kurtosis_threshold << "\n"
    << "mitigate_rfi_freq_list = " << mitigate_rfi_freq_list << "\n"
    << "signal_detect_signal_noise_threshold = "
    << signal_detect_signal_noise_threshold << "\n"
    << "signal_detect_max_boxcar_length = " << signal_detect_max_boxcar_length
    << "\n"
    << "input_file_path = " << input_file_path << "\n"
    << "input_file_offset_bytes = " << input_file_offset_bytes << "\n"
    << "baseband_output_file_prefix = " << baseband_output_file_prefix
    << "\n"
    << "log_level = " << log_level << "\n";
  return o.str();
}

}  // namespace srtb_app
