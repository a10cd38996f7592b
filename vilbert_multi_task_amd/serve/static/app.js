// ViLBERT multi-task demo client.
//
// Behavioral port of the reference's inline template JS, vanilla-JS
// (offline image: no jQuery / ReconnectingWebSocket / fileupload plugin):
//   - websocket lifecycle + {info,terminal,result} dispatch and per-task
//     result rendering       (result.html:32-262)
//   - image-count task gating, sample-image selection with FIFO eviction,
//     submit with GuessWhat regex validation (header.html:3-460)
//   - <=4-file upload flow                     (demo_images.html:64-128)
//   - task details fetch -> placeholder/example (result.html:4-27)
//
// Gating table (header.html:30-75): 1 image -> tasks 1/15/4/11/16/13
// (options 1-6); 2 images -> NLVR2 + retrieval (options 7-8); >2 images ->
// retrieval only (option 8); max 4 images, oldest evicted.

(function () {
  "use strict";
  var socketid = document.body.dataset.socketid;
  var $ = function (id) { return document.getElementById(id); };
  var sampleImagesList = [];
  var taskData = null;

  // ---- websocket with reconnect (result.html:32-36, onopen sends the
  // socketid — consumers.py:10-11 group subscription contract) ------------
  function connect() {
    var scheme = window.location.protocol === "https:" ? "wss" : "ws";
    var sock = new WebSocket(scheme + "://" + window.location.host + "/chat/");
    sock.onopen = function () { sock.send(socketid); };
    sock.onmessage = function (ev) { onPush(JSON.parse(ev.data)); };
    sock.onclose = function () { setTimeout(connect, 1000); };  // reconnect
    return sock;
  }
  connect();

  function terminalLine(text) {
    var li = document.createElement("li");
    li.textContent = text;
    $("comments").insertBefore(li, $("comments").firstChild);  // prepend
  }

  // ---- push dispatch (result.html:90-262) --------------------------------
  function onPush(response) {
    if ("terminal" in response) terminalLine(response.terminal);
    if (!("result" in response)) return;
    $("fileupload").disabled = false;
    $("submit-images").value = "Submit";
    $("task-example").innerHTML = "";
    $("task-example").style.display = "none";
    var result = JSON.parse(response.result);
    var tid = result.task_id;
    if (tid === "11" || tid === "16" || tid === "4" || tid === "7") {
      $("show-grad-cam-result").style.display = "none";
      var paths = [];
      var names = result.image_name_list || [];
      for (var i = 0; i < 4; i++) {
        if (names[i] === undefined) { paths.push(null); continue; }
        // grounding results live under media/refer_expressions_task/<uuid>.jpg
        // (worker.py:591-600); retrieval names are media-relative already
        paths.push(tid === "7" ? "/media/" + names[i]
                               : "/media/refer_expressions_task/" + names[i] + ".jpg");
      }
      for (var j = 0; j < 4; j++) {
        var img = $("ReferExpressionsTaskResultImage_" + (j + 1));
        var ans = $("ReferExpressionsTaskResultAnswer_" + (j + 1));
        if (paths[j]) {
          img.src = paths[j];
          ans.textContent = result.confidence_list[j] + "%";
          img.style.display = "";
          ans.style.display = "";
        } else {
          img.style.display = "none";
          ans.style.display = "none";
        }
      }
      $("show-refer-expressions-task-result").style.display = "";
    } else {
      $("show-refer-expressions-task-result").style.display = "none";
      var answers = result.result;
      var bars = tid === "12" ? 2 : 3;  // NLVR2 shows two bars (result.html:196)
      for (var b = 0; b < 3; b++) {
        var row = $("progress-bar-" + b);
        if (b >= bars) { row.style.display = "none"; continue; }
        row.style.display = "";
        var bar = $("progressbar" + b);
        bar.style.width = answers[b].confidence + "%";
        bar.setAttribute("aria-valuenow", answers[b].confidence);
        bar.textContent = answers[b].confidence.toFixed(2) + "%";
        $("predictedAnswer" + b).textContent = answers[b].answer;
      }
      $("show-grad-cam-result").style.display = "";
    }
  }

  // ---- task gating (header.html option-index table) ----------------------
  function setOptions(enabledIdx) {
    var opts = $("selected-task").options;
    for (var i = 1; i <= 8; i++) opts[i].disabled = enabledIdx.indexOf(i) < 0;
  }
  function gateByImageCount(n) {
    if (n === 1) setOptions([1, 2, 3, 4, 5, 6]);
    else if (n === 2) setOptions([7, 8]);
    else if (n > 2) setOptions([8]);
    else setOptions([]);
  }

  // ---- sample gallery (demo_images.html + header.html:3-90) --------------
  function resetTaskInputs() {
    $("task-example").innerHTML = "";
    $("task-example").style.display = "none";
    $("ResultDiv").style.display = "none";
    $("show-refer-expressions-task-result").style.display = "none";
    $("show-grad-cam-result").style.display = "none";
    $("question").value = "";
    $("selected-task").value = "";
    $("question").placeholder = "Input Field";
    $("submit-button").disabled = true;
  }

  function addImagesToSampleList(imgEl, imagePath) {
    resetTaskInputs();
    $("question").disabled = true;
    $("selected-task").disabled = true;
    $("fileupload").disabled = true;
    var idx = sampleImagesList.indexOf(imagePath);
    if (idx >= 0) {                      // deselect
      sampleImagesList.splice(idx, 1);
      imgEl.style.border = "";
      if (sampleImagesList.length === 0) $("fileupload").disabled = false;
    } else if (sampleImagesList.length < 4) {
      sampleImagesList.push(imagePath);
      imgEl.style.border = "3px solid red";
    } else {                             // full: evict oldest (header.html:76-82)
      var removed = sampleImagesList.splice(0, 1)[0];
      var old = document.querySelector('[data-path="' + removed + '"]');
      if (old) old.style.border = "";
      sampleImagesList.push(imagePath);
      imgEl.style.border = "3px solid red";
    }
  }

  function clearSubmittedImagesList() {
    sampleImagesList.forEach(function (p) {
      var el = document.querySelector('[data-path="' + p + '"]');
      if (el) el.style.border = "";
    });
    sampleImagesList = [];
    $("fileupload").disabled = false;
    resetTaskInputs();
  }

  // populate the gallery from the server-injected list (views.py:64-81)
  (window.DEMO_IMAGES || []).forEach(function (path) {
    var img = document.createElement("img");
    img.src = path;
    img.dataset.path = path;
    img.className = "demo-thumb";
    img.onclick = function () { addImagesToSampleList(img, path); };
    $("demo-image-gallery").appendChild(img);
  });

  // ---- submit selected/uploaded images into the result panel -------------
  function submitUploadedImage(srcList) {
    $("question").disabled = false;
    $("selected-task").disabled = false;
    $("ResultDiv").style.display = "";
    ["one-image-div", "two-images-div", "four-images-div"].forEach(function (d) {
      $(d).style.display = "none";
    });
    if (srcList.length === 1) {
      $("inputImageAfterUpload").src = srcList[0];
      $("one-image-div").style.display = "";
      terminalLine("Submitted demo image");
    } else if (srcList.length === 2) {
      $("inputImageAfterUpload_1").src = srcList[0];
      $("inputImageAfterUpload_2").src = srcList[1];
      $("two-images-div").style.display = "";
    } else {
      for (var i = 0; i < 4; i++) {
        var el = $("inputImageAfterUpload_" + (i + 1) + "4");
        if (srcList[i]) { el.src = srcList[i]; el.style.display = ""; }
        else { el.removeAttribute("src"); el.style.display = "none"; }
      }
      $("four-images-div").style.display = "";
    }
    gateByImageCount(srcList.length);
    sampleImagesList = [];
  }
  $("submit-images").onclick = function () {
    if (sampleImagesList.length) submitUploadedImage(sampleImagesList.slice());
  };
  $("clear-images").onclick = clearSubmittedImagesList;

  // ---- upload (<=4 files — demo_images.html:92-95) -----------------------
  $("fileupload").onchange = function () {
    var files = this.files;
    if (files.length > 4) {
      alert("Only a maximum of 4 files are allowed!");
      this.value = "";
      return;
    }
    var fd = new FormData();
    for (var i = 0; i < files.length; i++) fd.append("files[]", files[i]);
    $("progress-bar-div").style.display = "";
    fetch("/upload_image/", { method: "POST", body: fd })
      .then(function (r) { return r.json(); })
      .then(function (data) {
        $("progress-bar-div").style.display = "none";
        submitUploadedImage(data.file_paths);
      });
  };

  // ---- task dropdown -> details fetch (result.html:4-27) -----------------
  $("selected-task").onchange = function () {
    var tid = this.value;
    $("question").value = "";
    $("task-example").innerHTML = "";
    $("task-example").style.display = "none";
    fetch("/get_task_details/" + tid + "/")
      .then(function (r) { return r.json(); })
      .then(function (data) {
        taskData = data;
        $("question").placeholder = data.placeholder || "Input Field";
        if (data.example) {
          $("task-example").textContent = data.example;
          $("task-example").style.display = "";
        }
        updateSubmitEnabled();
      });
  };
  function updateSubmitEnabled() {
    $("submit-button").disabled = !($("question").value !== "" &&
                                    $("selected-task").value !== "");
  }
  $("question").onkeyup = updateSubmitEnabled;

  // ---- final submit (header.html:374-459) --------------------------------
  function currentImageList(taskId) {
    function pathOf(el) { return new URL(el.src, window.location).pathname; }
    if (taskId === "12") {
      return [pathOf($("inputImageAfterUpload_1")), pathOf($("inputImageAfterUpload_2"))];
    }
    if (taskId === "7") {
      var out = [];
      for (var i = 0; i < 4; i++) {
        var el = $("inputImageAfterUpload_" + (i + 1) + "4");
        if (el.getAttribute("src")) out.push(pathOf(el));
      }
      return out;
    }
    return [pathOf($("inputImageAfterUpload"))];
  }
  $("submit-button").onclick = function () {
    $("show-grad-cam-result").style.display = "none";
    $("show-refer-expressions-task-result").style.display = "none";
    var question = $("question").value;
    var taskId = $("selected-task").value;
    if (!taskId) {
      alert("The number selected images and task type doesn't match. " +
            "Please select the appropriate task type from the dropdown.");
      return;
    }
    if (taskId === "16") {
      // GuessWhat dialog validation (header.html:380-396) — same regex
      var re = /^(Q:[a-z\d\-_\s]+\?\sA:[a-z\d\-_\s]+)$/;
      var parts = question.split(",");
      for (var i = 0; i < parts.length; i++) {
        if (!re.test(parts[i].trim())) {
          alert("The input " + parts[i] + " to the task is wrong. Please enter in below format. \r\nQ: 1st question? A: 1st answer, Q: 2nd question? A: 2nd answer, ... \r\n E.g. Q: is it white? A: no, Q: is it red? A: yes, Q: is it shiny? A: yes");
          return;
        }
      }
      question = question.replace(",", "");  // reference replaces FIRST comma only
    }
    var imgs = currentImageList(taskId);
    var body = new URLSearchParams();
    imgs.forEach(function (p) { body.append("image_list[]", p); });
    body.append("question", question);
    body.append("socket_id", socketid);
    body.append("task_id", taskId);
    fetch("/", { method: "POST", body: body });
    sampleImagesList = [];
  };
})();
