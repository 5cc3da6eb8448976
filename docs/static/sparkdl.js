// sparkdl API docs helper (original): collapse long parameter lists and
// add anchor affordances on hover.
document.addEventListener("DOMContentLoaded", function () {
  document.querySelectorAll("dl.py > dt").forEach(function (dt) {
    dt.addEventListener("mouseenter", function () {
      dt.style.cursor = "pointer";
    });
    dt.addEventListener("click", function () {
      var dd = dt.nextElementSibling;
      if (dd && dd.tagName === "DD") {
        dd.style.display = dd.style.display === "none" ? "" : "none";
      }
    });
  });
});
